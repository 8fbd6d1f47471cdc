"""Dreamer-V3 training loop.

Parity: sheeprl/algos/dreamer_v3/dreamer_v3.py — train :48 (dynamic learning
:113-145, imagination :235-241, actor loss :262-305, critic loss :307-325),
main :361 (env interaction :553-657, replay-ratio-driven training :664-680,
EMA target update :678-680, checkpointing :737-763).

MI355X notes:
* sequence batch [T, B] trains in bf16-true; stochastic-state categoricals
  and KLs run in fp32 (loss.py).
* replay samples are gathered on host then moved with non_blocking pinned
  copies; the heavy kernels (GRU gates, LN+SiLU, λ-scan, Adam) are the fused
  HIP ops.
* the EMA target-critic update is one multi-tensor kernel (ops.ema_update_).
"""

from __future__ import annotations

import os
from typing import Any, Dict, Sequence

import numpy as np
import torch
import torch.distributions as td
import torch.nn.functional as F

from sheeprl_amd import ops
from sheeprl_amd.algos.dreamer_v3.agent import build_agent
from sheeprl_amd.algos.dreamer_v3.imagine import imagine_applicable, imagine_rollout
from sheeprl_amd.algos.dreamer_v3.loss import reconstruction_loss, reconstruction_loss_forked
from sheeprl_amd.algos.dreamer_v3.utils import (
    AGGREGATOR_KEYS,
    Moments,
    compute_lambda_values,
    prepare_obs,
    test,
)
from sheeprl_amd.config import save_config
from sheeprl_amd.data import EnvIndependentReplayBuffer, EpisodeBuffer, SequentialReplayBuffer
from sheeprl_amd.distributions import (
    BernoulliSafeMode,
    MSEDistribution,
    SymlogDistribution,
    TwoHotEncodingDistribution,
)
from sheeprl_amd.envs import make_env, spaces, vectorize_env
from sheeprl_amd.optim import make_optimizer
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.parallel.streams import Branches
from sheeprl_amd.utils.logger import get_log_dir, get_logger
from sheeprl_amd.utils.metric import MetricAggregator
from sheeprl_amd.utils.registry import register_algorithm
from sheeprl_amd.utils.timer import timer
from sheeprl_amd.utils.utils import Ratio


def train(
    runtime: Runtime,
    world_model: Any,
    actor: Any,
    critic: Any,
    target_critic: Any,
    world_optimizer: torch.optim.Optimizer,
    actor_optimizer: torch.optim.Optimizer,
    critic_optimizer: torch.optim.Optimizer,
    data: Dict[str, torch.Tensor],
    aggregator: MetricAggregator,
    cfg: Any,
    is_continuous: bool,
    actions_dim: Sequence[int],
    moments: Moments,
    metrics_out: Dict[str, torch.Tensor] = None,
) -> None:
    batch_size = cfg.algo.per_rank_batch_size
    sequence_length = cfg.algo.per_rank_sequence_length
    recurrent_state_size = cfg.algo.world_model.recurrent_model.recurrent_state_size
    stochastic_size = cfg.algo.world_model.stochastic_size
    discrete_size = cfg.algo.world_model.discrete_size
    stoch_state_size = stochastic_size * discrete_size
    device = runtime.device
    dtype = runtime.param_dtype

    batch_obs = {k: ops.normalize_obs(data[k]).to(dtype) for k in cfg.algo.cnn_keys.encoder}
    batch_obs.update({k: data[k].to(dtype) for k in cfg.algo.mlp_keys.encoder})
    data["is_first"][0, :] = torch.ones_like(data["is_first"][0, :])

    # shift actions right by one (a_0 := 0) — the stored action at t led to obs t+1
    batch_actions = torch.cat((torch.zeros_like(data["actions"][:1]), data["actions"][:-1]), dim=0)

    # ---------------- dynamic learning ----------------
    recurrent_state = torch.zeros(1, batch_size, recurrent_state_size, device=device, dtype=dtype)
    recurrent_states = torch.empty(sequence_length, batch_size, recurrent_state_size, device=device, dtype=dtype)
    priors_logits = torch.empty(sequence_length, batch_size, stoch_state_size, device=device)

    embedded_obs = world_model.encoder(batch_obs)

    if cfg.algo.world_model.decoupled_rssm:
        posteriors_logits, posteriors = world_model.rssm._representation(embedded_obs)
        for i in range(sequence_length):
            posterior = torch.zeros_like(posteriors[:1]) if i == 0 else posteriors[i - 1 : i]
            recurrent_state, _, prior_logits = world_model.rssm.dynamic(
                posterior, recurrent_state, batch_actions[i : i + 1], data["is_first"][i : i + 1]
            )
            recurrent_states[i] = recurrent_state
            priors_logits[i] = prior_logits
    else:
        # initial states are step-invariant: hoist them out of the scan; the
        # prior (transition) head batches over all T afterwards
        _ir, _ip = world_model.rssm.get_initial_states((1, batch_size))
        initial_states = (_ir.contiguous(), _ip.contiguous())
        from sheeprl_amd.ops.scan import rssm_scan, scan_applicable

        if device.type == "cuda" and cfg.algo.get("fused_scan", True) and scan_applicable(world_model.rssm):
            # fused scan: hand-written backward, no per-step autograd overhead
            recurrent_states, posteriors_flat, posteriors_logits = rssm_scan(
                world_model.rssm, embedded_obs, batch_actions, data["is_first"], initial_states
            )
            posteriors = posteriors_flat.view(sequence_length, batch_size, stochastic_size, discrete_size)
        else:
            posterior = torch.zeros(1, batch_size, stochastic_size, discrete_size, device=device, dtype=dtype)
            posteriors = torch.empty(
                sequence_length, batch_size, stochastic_size, discrete_size, device=device, dtype=dtype
            )
            posteriors_logits = torch.empty(sequence_length, batch_size, stoch_state_size, device=device)
            for i in range(sequence_length):
                recurrent_state, posterior, posterior_logits = world_model.rssm.dynamic_posterior(
                    posterior,
                    recurrent_state,
                    batch_actions[i : i + 1],
                    embedded_obs[i : i + 1],
                    data["is_first"][i : i + 1],
                    initial_states=initial_states,
                )
                recurrent_states[i] = recurrent_state
                posteriors[i] = posterior
                posteriors_logits[i] = posterior_logits
        # the batched prior (transition) head is off the scan's critical
        # path; it is computed at the loss site (inside a stream fork on GPU)
        priors_logits = None
    latent_states = torch.cat((posteriors.view(*posteriors.shape[:-2], -1), recurrent_states), -1)

    # Intra-graph concurrency (parallel/streams.py): the world-model head
    # chains and the actor/critic behaviour phases are mutually independent;
    # forking them onto side streams lets the captured hipGraph replay run
    # them concurrently on idle CUs.  Enabled on single-rank GPU runs only —
    # multi-rank backward interleaves GradSync collectives, which stay
    # single-stream for deterministic enqueue order.
    # region selection for A/B: "0"=off, "all", "heads" (world-model head
    # forks only), "behaviour" (actor/critic + trajectory-sweep forks only)
    _sm = os.environ.get("SHEEPRL_AMD_GRAPH_STREAMS", "0")
    _stream_ok = device.type == "cuda" and runtime.world_size == 1 and bool(cfg.algo.get("graph_streams", True))
    br = Branches(_stream_ok and _sm in ("1", "all", "heads"))
    br_beh = Branches(_stream_ok and _sm in ("1", "all", "behaviour"))

    # on the HIP path the fused NLL kernels take bf16 predictions directly —
    # skipping the fp32 upcast halves the loss-section reads and removes two
    # cast launches per key (fwd + backward grad cast)
    _no_cast = device.type == "cuda" and ops.use_hip(latent_states)
    continues_targets = 1 - data["terminated"]

    world_optimizer.zero_grad(set_to_none=True)
    if br.enabled:
        (
            rec_loss,
            kl,
            state_loss,
            reward_loss,
            observation_loss,
            continue_loss,
            priors_logits,
            posteriors_logits,
        ) = reconstruction_loss_forked(
            br,
            world_model,
            latent_states,
            recurrent_states,
            batch_obs,
            data["rewards"],
            continues_targets,
            posteriors_logits,
            priors_logits,
            stochastic_size,
            discrete_size,
            cfg.algo.cnn_keys.decoder,
            cfg.algo.mlp_keys.decoder,
            cfg.algo.world_model.kl_dynamic,
            cfg.algo.world_model.kl_representation,
            cfg.algo.world_model.kl_free_nats,
            cfg.algo.world_model.kl_regularizer,
            cfg.algo.world_model.continue_scale_factor,
            _no_cast,
        )
    else:
        if priors_logits is None:
            priors_logits = world_model.rssm.transition_logits(recurrent_states)
        reconstructed_obs = world_model.observation_model(latent_states)
        _c = (lambda t: t) if _no_cast else (lambda t: t.float())
        po = {
            k: MSEDistribution(_c(reconstructed_obs[k]), dims=len(reconstructed_obs[k].shape[2:]))
            for k in cfg.algo.cnn_keys.decoder
        }
        po.update(
            {
                k: SymlogDistribution(_c(reconstructed_obs[k]), dims=len(reconstructed_obs[k].shape[2:]))
                for k in cfg.algo.mlp_keys.decoder
            }
        )
        pr = TwoHotEncodingDistribution(world_model.reward_model(latent_states).float(), dims=1)
        pc = td.Independent(BernoulliSafeMode(logits=world_model.continue_model(latent_states).float()), 1)

        priors_logits = priors_logits.view(*priors_logits.shape[:-1], stochastic_size, discrete_size)
        posteriors_logits = posteriors_logits.view(*posteriors_logits.shape[:-1], stochastic_size, discrete_size)

        rec_loss, kl, state_loss, reward_loss, observation_loss, continue_loss = reconstruction_loss(
            po,
            {k: v.float() for k, v in batch_obs.items()},
            pr,
            data["rewards"],
            priors_logits,
            posteriors_logits,
            cfg.algo.world_model.kl_dynamic,
            cfg.algo.world_model.kl_representation,
            cfg.algo.world_model.kl_free_nats,
            cfg.algo.world_model.kl_regularizer,
            pc,
            continues_targets,
            cfg.algo.world_model.continue_scale_factor,
        )
    runtime.backward(rec_loss)
    world_model_grads = None
    if cfg.algo.world_model.clip_gradients and cfg.algo.world_model.clip_gradients > 0:
        world_model_grads = runtime.clip_gradients(world_model, world_optimizer, cfg.algo.world_model.clip_gradients)
    world_optimizer.step()

    # ---------------- behaviour learning (imagination) ----------------
    horizon = cfg.algo.horizon
    flat = batch_size * sequence_length
    # for the discrete (REINFORCE) actor nothing backpropagates through the
    # rollout, so on GPU it runs as pure inference with the launch-lean
    # buffer-reusing rollout (imagine.py); the module loop remains the
    # continuous-actions / CPU / non-canonical-architecture path
    use_fast_imagine = (
        cfg.algo.get("fused_imagination", True)
        and not is_continuous
        and device.type == "cuda"
        and imagine_applicable(world_model.rssm, actor)
    )
    if use_fast_imagine:
        imagined_trajectories, imagined_actions = imagine_rollout(
            world_model.rssm,
            actor,
            posteriors.detach().reshape(flat, stoch_state_size).to(dtype),
            recurrent_states.detach().reshape(flat, recurrent_state_size).to(dtype),
            horizon,
        )
    else:
        imagined_prior = posteriors.detach().reshape(1, -1, stoch_state_size)
        recurrent_state = recurrent_states.detach().reshape(1, -1, recurrent_state_size)
        imagined_latent_state = torch.cat((imagined_prior, recurrent_state), -1)
        imagined_trajectories = torch.empty(
            horizon + 1, flat, stoch_state_size + recurrent_state_size, device=device, dtype=dtype
        )
        imagined_trajectories[0] = imagined_latent_state
        imagined_actions = torch.empty(horizon + 1, flat, data["actions"].shape[-1], device=device, dtype=dtype)
        actions = torch.cat(actor(imagined_latent_state.detach())[0], dim=-1).to(dtype)
        imagined_actions[0] = actions

        for i in range(1, horizon + 1):
            imagined_prior, recurrent_state = world_model.rssm.imagination(imagined_prior, recurrent_state, actions)
            imagined_prior = imagined_prior.view(1, -1, stoch_state_size).to(dtype)
            imagined_latent_state = torch.cat((imagined_prior, recurrent_state), -1)
            imagined_trajectories[i] = imagined_latent_state
            actions = torch.cat(actor(imagined_latent_state.detach())[0], dim=-1).to(dtype)
            imagined_actions[i] = actions

    # fast behaviour-loss path (discrete single-head on HIP): for a REINFORCE
    # actor nothing backpropagates through the imagined values/returns, so the
    # value/reward/continue heads run inference-only with fused twohot means,
    # and each loss is ONE kernel per direction instead of the autograd chain
    fast_losses = use_fast_imagine and len(actions_dim) == 1
    if fast_losses:
        with torch.no_grad():
            # the three trajectory sweeps are independent — fork them
            with br_beh.fork():
                predicted_values = ops.twohot_mean(critic(imagined_trajectories))
            with br_beh.fork():
                predicted_rewards = ops.twohot_mean(world_model.reward_model(imagined_trajectories))
            with br_beh.fork():
                continues = (world_model.continue_model(imagined_trajectories) > 0).float()
            br_beh.join()
            true_continue = (1 - data["terminated"]).flatten().reshape(1, -1, 1)
            continues = torch.cat((true_continue, continues[1:]))
            lambda_values = compute_lambda_values(
                predicted_rewards[1:], predicted_values[1:], continues[1:] * cfg.algo.gamma,
                lmbda=cfg.algo.lmbda,
            )
            discount = torch.cumprod(continues * cfg.algo.gamma, dim=0) / cfg.algo.gamma

        # Moments mutates EMA state and (multi-rank) all-gathers: keep it on
        # the ambient stream, ahead of the actor/critic forks
        offset, invscale = moments(lambda_values, runtime)

        # the whole actor phase and the whole critic phase (forward, backward,
        # clip, optimizer step) touch disjoint parameters and read-only share
        # the trajectories/λ-values: run them concurrently
        with br_beh.fork():
            # actor loss (advantage offsets cancel: (λ-off)/s - (v-off)/s = (λ-v)/s)
            actor_optimizer.zero_grad(set_to_none=True)
            policies = actor(imagined_trajectories.detach())[1]
            with torch.no_grad():
                advantage = (lambda_values - predicted_values[:-1]) / invscale
            policy_loss = ops.reinforce_loss(
                policies[0].logits, imagined_actions, advantage, discount[:-1], cfg.algo.actor.ent_coef
            )
            runtime.backward(policy_loss)
            actor_grads = None
            if cfg.algo.actor.clip_gradients and cfg.algo.actor.clip_gradients > 0:
                actor_grads = runtime.clip_gradients(actor, actor_optimizer, cfg.algo.actor.clip_gradients)
            actor_optimizer.step()

        with br_beh.fork():
            # critic loss: two two-hot CEs over shared logits in one kernel
            critic_optimizer.zero_grad(set_to_none=True)
            qv_logits = critic(imagined_trajectories.detach()[:-1]).float()
            with torch.no_grad():
                predicted_target_values = ops.twohot_mean(target_critic(imagined_trajectories.detach()[:-1]))
            value_loss = ops.critic_twohot_loss(
                qv_logits, lambda_values, predicted_target_values, discount[:-1]
            )
            runtime.backward(value_loss)
            critic_grads = None
            if cfg.algo.critic.clip_gradients and cfg.algo.critic.clip_gradients > 0:
                critic_grads = runtime.clip_gradients(critic, critic_optimizer, cfg.algo.critic.clip_gradients)
            critic_optimizer.step()
        br_beh.join()
    else:
        predicted_values = TwoHotEncodingDistribution(critic(imagined_trajectories).float(), dims=1).mean
        predicted_rewards = TwoHotEncodingDistribution(world_model.reward_model(imagined_trajectories).float(), dims=1).mean
        continues = td.Independent(
            BernoulliSafeMode(logits=world_model.continue_model(imagined_trajectories).float()), 1
        ).mode
        true_continue = (1 - data["terminated"]).flatten().reshape(1, -1, 1)
        continues = torch.cat((true_continue, continues[1:]))

        lambda_values = compute_lambda_values(
            predicted_rewards[1:], predicted_values[1:], continues[1:] * cfg.algo.gamma, lmbda=cfg.algo.lmbda
        )

        with torch.no_grad():
            discount = torch.cumprod(continues * cfg.algo.gamma, dim=0) / cfg.algo.gamma

        # actor loss
        actor_optimizer.zero_grad(set_to_none=True)
        policies = actor(imagined_trajectories.detach())[1]
        baseline = predicted_values[:-1]
        offset, invscale = moments(lambda_values, runtime)
        normed_lambda_values = (lambda_values - offset) / invscale
        normed_baseline = (baseline - offset) / invscale
        advantage = normed_lambda_values - normed_baseline
        if is_continuous:
            objective = advantage
        else:
            objective = (
                torch.stack(
                    [
                        p.log_prob(imgnd_act.detach().float()).unsqueeze(-1)[:-1]
                        for p, imgnd_act in zip(policies, torch.split(imagined_actions, list(actions_dim), dim=-1))
                    ],
                    dim=-1,
                ).sum(dim=-1)
                * advantage.detach()
            )
        try:
            entropy = cfg.algo.actor.ent_coef * torch.stack([p.entropy() for p in policies], -1).sum(dim=-1)
        except NotImplementedError:
            entropy = torch.zeros_like(objective)
        policy_loss = -torch.mean(discount[:-1].detach() * (objective + entropy.unsqueeze(dim=-1)[:-1]))
        runtime.backward(policy_loss)
        actor_grads = None
        if cfg.algo.actor.clip_gradients and cfg.algo.actor.clip_gradients > 0:
            actor_grads = runtime.clip_gradients(actor, actor_optimizer, cfg.algo.actor.clip_gradients)
        actor_optimizer.step()

        # critic loss (with EMA-critic regularizer)
        qv = TwoHotEncodingDistribution(critic(imagined_trajectories.detach()[:-1]).float(), dims=1)
        predicted_target_values = TwoHotEncodingDistribution(
            target_critic(imagined_trajectories.detach()[:-1]).float(), dims=1
        ).mean

        critic_optimizer.zero_grad(set_to_none=True)
        value_loss = -qv.log_prob(lambda_values.detach())
        value_loss = value_loss - qv.log_prob(predicted_target_values.detach())
        value_loss = torch.mean(value_loss * discount[:-1].squeeze(-1))
        runtime.backward(value_loss)
        critic_grads = None
        if cfg.algo.critic.clip_gradients and cfg.algo.critic.clip_gradients > 0:
            critic_grads = runtime.clip_gradients(critic, critic_optimizer, cfg.algo.critic.clip_gradients)
        critic_optimizer.step()

    if metrics_out is not None:
        # under hipGraph capture these are STATIC buffers: stashing the
        # references once lets the main loop feed the aggregator after every
        # replay instead of running a ~14x-slower eager step for metrics
        metrics_out.update(
            rec_loss=rec_loss.detach(),
            observation_loss=observation_loss.detach(),
            reward_loss=reward_loss.detach(),
            state_loss=state_loss.detach(),
            continue_loss=continue_loss.detach(),
            kl=kl.detach(),
            posteriors_logits=posteriors_logits.detach(),
            priors_logits=priors_logits.detach(),
            policy_loss=policy_loss.detach(),
            value_loss=value_loss.detach(),
        )
        if world_model_grads is not None:
            metrics_out["world_model_grads"] = world_model_grads.detach()
        if actor_grads is not None:
            metrics_out["actor_grads"] = actor_grads.detach()
        if critic_grads is not None:
            metrics_out["critic_grads"] = critic_grads.detach()

    if aggregator and not MetricAggregator.disabled:
        aggregator.update("Loss/world_model_loss", rec_loss.detach())
        aggregator.update("Loss/observation_loss", observation_loss.detach())
        aggregator.update("Loss/reward_loss", reward_loss.detach())
        aggregator.update("Loss/state_loss", state_loss.detach())
        aggregator.update("Loss/continue_loss", continue_loss.detach())
        aggregator.update("State/kl", kl.mean().detach())
        aggregator.update(
            "State/post_entropy",
            td.Independent(td.OneHotCategorical(logits=posteriors_logits.detach().float()), 1).entropy().mean(),
        )
        aggregator.update(
            "State/prior_entropy",
            td.Independent(td.OneHotCategorical(logits=priors_logits.detach().float()), 1).entropy().mean(),
        )
        aggregator.update("Loss/policy_loss", policy_loss.detach())
        aggregator.update("Loss/value_loss", value_loss.detach())
        if world_model_grads is not None:
            aggregator.update("Grads/world_model", world_model_grads.detach())
        if actor_grads is not None:
            aggregator.update("Grads/actor", actor_grads.detach())
        if critic_grads is not None:
            aggregator.update("Grads/critic", critic_grads.detach())

    actor_optimizer.zero_grad(set_to_none=True)
    critic_optimizer.zero_grad(set_to_none=True)
    world_optimizer.zero_grad(set_to_none=True)


def _capture_train_step(
    runtime, world_model, actor, critic, target_critic,
    world_optimizer, actor_optimizer, critic_optimizer,
    example_batch, cfg, is_continuous, actions_dim, moments,
):
    """hipGraph-capture the gradient step; returns a replayable callable or
    None when capture fails (falls back to eager)."""
    from sheeprl_amd.parallel.graphs import CUDAGraphStep
    from sheeprl_amd.utils.metric import MetricAggregator

    metrics_out: Dict[str, torch.Tensor] = {}

    def train_fn(batch):
        was_disabled = MetricAggregator.disabled
        MetricAggregator.disabled = True
        try:
            metrics_out.clear()
            train(
                runtime, world_model, actor, critic, target_critic,
                world_optimizer, actor_optimizer, critic_optimizer,
                batch, None, cfg, is_continuous, actions_dim, moments,
                metrics_out=metrics_out,
            )
        finally:
            MetricAggregator.disabled = was_disabled

    try:
        # NOTE: warmup (2) + capture (1) each execute a real gradient step;
        # these three updates are not counted by the cumulative counters.
        # metrics_out holds references to the capture's STATIC loss/grad
        # tensors: after each replay they carry the replayed batch's values.
        step = CUDAGraphStep(train_fn, example_batch, warmup=2)
        runtime.print("[dreamer_v3] gradient step captured in a hipGraph")
        return step, metrics_out
    except Exception as e:  # noqa: BLE001
        runtime.print(f"[dreamer_v3] hipGraph capture failed ({e}); eager training")
        return None, None


def _feed_graph_metrics(aggregator: MetricAggregator, m: Dict[str, torch.Tensor]) -> None:
    """Feed the aggregator from the captured graph's static loss/grad
    buffers after a replay.  Clones are device-side (ordered after the
    replay on the same stream); the only host syncs remain at
    ``aggregator.compute()`` on the log cadence — matching the reference's
    per-train-call metric updates without eager metric steps."""
    aggregator.update("Loss/world_model_loss", m["rec_loss"].clone())
    aggregator.update("Loss/observation_loss", m["observation_loss"].clone())
    aggregator.update("Loss/reward_loss", m["reward_loss"].clone())
    aggregator.update("Loss/state_loss", m["state_loss"].clone())
    aggregator.update("Loss/continue_loss", m["continue_loss"].clone())
    aggregator.update("State/kl", m["kl"].mean())
    aggregator.update(
        "State/post_entropy",
        td.Independent(td.OneHotCategorical(logits=m["posteriors_logits"].float()), 1).entropy().mean(),
    )
    aggregator.update(
        "State/prior_entropy",
        td.Independent(td.OneHotCategorical(logits=m["priors_logits"].float()), 1).entropy().mean(),
    )
    aggregator.update("Loss/policy_loss", m["policy_loss"].clone())
    aggregator.update("Loss/value_loss", m["value_loss"].clone())
    if "world_model_grads" in m:
        aggregator.update("Grads/world_model", m["world_model_grads"].clone())
    if "actor_grads" in m:
        aggregator.update("Grads/actor", m["actor_grads"].clone())
    if "critic_grads" in m:
        aggregator.update("Grads/critic", m["critic_grads"].clone())


@register_algorithm(name="dreamer_v3")
def main(runtime: Runtime, cfg: Any) -> None:
    device = runtime.device
    dtype = runtime.param_dtype

    log_dir = get_log_dir(runtime, cfg.root_dir, cfg.run_name)
    logger = get_logger(runtime, cfg, log_dir)
    runtime.logger = logger
    if runtime.is_global_zero:
        save_config(cfg, os.path.join(log_dir, "config.yaml"))

    envs = vectorize_env(cfg, cfg.seed, runtime.global_rank)
    obs_space = envs.single_observation_space
    action_space = envs.single_action_space
    is_continuous = isinstance(action_space, spaces.Box)
    is_multidiscrete = isinstance(action_space, spaces.MultiDiscrete)
    actions_dim = tuple(
        action_space.shape if is_continuous else (action_space.nvec.tolist() if is_multidiscrete else [action_space.n])
    )
    cnn_keys = list(cfg.algo.cnn_keys.encoder or [])
    mlp_keys = list(cfg.algo.mlp_keys.encoder or [])
    obs_keys = cnn_keys + mlp_keys

    state: Dict[str, Any] = {}
    if cfg.checkpoint.resume_from:
        state = runtime.load(cfg.checkpoint.resume_from)

    world_model, actor, critic, target_critic, player = build_agent(
        runtime,
        actions_dim,
        is_continuous,
        cfg,
        obs_space,
        state.get("world_model"),
        state.get("actor"),
        state.get("critic"),
        state.get("target_critic"),
    )

    world_optimizer = make_optimizer(world_model.parameters(), cfg.algo.world_model.optimizer)
    actor_optimizer = make_optimizer(actor.parameters(), cfg.algo.actor.optimizer)
    critic_optimizer = make_optimizer(critic.parameters(), cfg.algo.critic.optimizer)
    if "world_optimizer" in state:
        world_optimizer.load_state_dict(state["world_optimizer"])
        actor_optimizer.load_state_dict(state["actor_optimizer"])
        critic_optimizer.load_state_dict(state["critic_optimizer"])

    moments = Moments(
        cfg.algo.actor.moments.decay,
        cfg.algo.actor.moments.max,
        cfg.algo.actor.moments.percentile.low,
        cfg.algo.actor.moments.percentile.high,
    ).to(device)
    if "moments" in state:
        moments.load_state_dict(state["moments"])

    aggregator = MetricAggregator(
        {k: "mean" for k in AGGREGATOR_KEYS}, sync_on_compute=cfg.metric.get("sync_on_compute", False)
    )

    # buffer: env-independent sequential (dreamer_v3.py:479-485) or episode
    buffer_size = max(int(cfg.buffer.size), 1)
    if cfg.buffer.get("type", "sequential") == "episode":
        rb: Any = EpisodeBuffer(
            buffer_size,
            sequence_length=cfg.algo.per_rank_sequence_length,
            n_envs=cfg.env.num_envs,
            obs_keys=obs_keys,
            prioritize_ends=cfg.buffer.get("prioritize_ends", False),
            memmap=cfg.buffer.memmap,
            memmap_dir=os.path.join(log_dir, "memmap_buffer", f"rank_{runtime.global_rank}"),
        )
    else:
        rb = EnvIndependentReplayBuffer(
            buffer_size,
            n_envs=cfg.env.num_envs,
            obs_keys=obs_keys,
            memmap=cfg.buffer.memmap,
            memmap_dir=os.path.join(log_dir, "memmap_buffer", f"rank_{runtime.global_rank}"),
            buffer_cls=SequentialReplayBuffer,
        )
    if "rb" in state and state["rb"] is not None:
        rbs = state["rb"]
        if isinstance(rbs, list):
            rb.load_state_dict(rbs[runtime.global_rank % len(rbs)])
        else:
            rb.load_state_dict(rbs)

    # counters (parity: work_with_steps.md)
    world_size = runtime.world_size
    num_envs = cfg.env.num_envs
    policy_steps_per_iter = int(num_envs * world_size)
    total_iters = int(cfg.algo.total_steps // policy_steps_per_iter) if not cfg.dry_run else 1
    learning_starts = cfg.algo.learning_starts // policy_steps_per_iter if not cfg.dry_run else 0
    # replay-ratio accounting starts AFTER the prefill (parity:
    # sheeprl dreamer_v3.py:661, sac.py:301 — the reference subtracts the
    # prefill policy steps before asking Ratio how many grad steps are owed,
    # otherwise the first train iteration pays a learning_starts-sized backlog)
    prefill_steps = max(learning_starts - 1, 0) * policy_steps_per_iter
    if cfg.checkpoint.resume_from:
        cfg.algo.per_rank_batch_size = state["batch_size"] // world_size
    start_iter = int(state.get("iter_num", 1))
    policy_step = int(state.get("policy_step", (start_iter - 1) * policy_steps_per_iter))
    last_log = int(state.get("last_log", 0))
    last_checkpoint = int(state.get("last_checkpoint", 0))

    ratio = Ratio(cfg.algo.replay_ratio, pretrain_steps=cfg.algo.per_rank_pretrain_steps)
    if "ratio" in state:
        ratio.load_state_dict(state["ratio"])

    clip_rewards_fn = (lambda r: np.tanh(r)) if cfg.env.clip_rewards else (lambda r: r)

    want_graphs = (
        bool(cfg.algo.get("hip_graphs", True))
        and (device.type == "cuda" or os.environ.get("SHEEPRL_AMD_FORCE_GRAPHS") == "1")
        and os.environ.get("SHEEPRL_AMD_NO_GRAPHS", "0") != "1"
    )
    graphed_step = None
    graph_metrics = None

    # initial step data
    step_data: Dict[str, np.ndarray] = {}
    obs, _ = envs.reset(seed=cfg.seed + runtime.global_rank * num_envs)
    for k in obs_keys:
        step_data[k] = np.asarray(obs[k])[np.newaxis]
    step_data["rewards"] = np.zeros((1, num_envs, 1), dtype=np.float32)
    step_data["truncated"] = np.zeros((1, num_envs, 1), dtype=np.float32)
    step_data["terminated"] = np.zeros((1, num_envs, 1), dtype=np.float32)
    step_data["is_first"] = np.ones_like(step_data["terminated"])
    player.init_states()

    cumulative_per_rank_gradient_steps = 0
    for iter_num in range(start_iter, total_iters + 1):
        policy_step += policy_steps_per_iter

        with torch.inference_mode():
            with timer("Time/env_interaction_time"):
                if iter_num <= learning_starts and cfg.checkpoint.resume_from is None:
                    real_actions = actions = np.stack([envs.single_action_space.sample() for _ in range(num_envs)])
                    if not is_continuous:
                        actions = np.concatenate(
                            [
                                F.one_hot(torch.as_tensor(a).long(), d).numpy()
                                for a, d in zip(np.asarray(actions).reshape(num_envs, -1).T, actions_dim)
                            ],
                            axis=-1,
                        ).reshape(num_envs, -1)
                else:
                    torch_obs = prepare_obs(runtime, obs, cnn_keys=cnn_keys, num_envs=num_envs)
                    acts = player.get_actions(torch_obs)
                    actions = torch.cat(acts, -1).view(num_envs, -1).float().cpu().numpy()
                    if is_continuous:
                        real_actions = actions
                    else:
                        real_actions = (
                            torch.stack([a.argmax(dim=-1) for a in acts], dim=-1).view(num_envs, -1).cpu().numpy()
                        )
                        if real_actions.shape[-1] == 1:
                            real_actions = real_actions[..., 0]

                step_data["actions"] = np.asarray(actions, dtype=np.float32).reshape(1, num_envs, -1)
                rb.add(step_data)

                next_obs, rewards, terminated, truncated, infos = envs.step(real_actions)
                dones = np.logical_or(terminated, truncated)

            step_data["is_first"] = np.zeros_like(step_data["terminated"])
            if any(infos.get("restart_on_exception", [])):
                for i, roe in enumerate(infos["restart_on_exception"]):
                    if roe and not dones[i]:
                        sub = rb.buffer[i] if hasattr(rb, "buffer") and isinstance(rb.buffer, list) else None
                        if sub is not None and len(sub) > 0:
                            last_idx = (sub._pos - 1) % sub.buffer_size
                            sub._buf["terminated"][last_idx] = 0.0
                            sub._buf["truncated"][last_idx] = 1.0
                            sub._buf["is_first"][last_idx] = 0.0
                        step_data["is_first"][:, i] = 1.0

            if cfg.metric.log_level > 0:
                for i, ep in enumerate(infos.get("episode", [])):
                    if ep is not None:
                        aggregator.update("Rewards/rew_avg", float(ep["r"][0]))
                        aggregator.update("Game/ep_len_avg", float(ep["l"][0]))

            real_next_obs = {k: np.asarray(v).copy() for k, v in next_obs.items()}
            for idx, final_obs in enumerate(infos.get("final_observation", [])):
                if final_obs is not None:
                    for k in obs_keys:
                        real_next_obs[k][idx] = final_obs[k]

            for k in obs_keys:
                step_data[k] = np.asarray(next_obs[k])[np.newaxis]
            obs = next_obs

            rewards = np.asarray(rewards, dtype=np.float32).reshape(1, num_envs, 1)
            step_data["terminated"] = np.asarray(terminated, dtype=np.float32).reshape(1, num_envs, 1)
            step_data["truncated"] = np.asarray(truncated, dtype=np.float32).reshape(1, num_envs, 1)
            step_data["rewards"] = clip_rewards_fn(rewards)

            dones_idxes = np.nonzero(dones)[0].tolist()
            if dones_idxes:
                reset_data = {}
                for k in obs_keys:
                    reset_data[k] = real_next_obs[k][dones_idxes][np.newaxis]
                reset_data["terminated"] = step_data["terminated"][:, dones_idxes]
                reset_data["truncated"] = step_data["truncated"][:, dones_idxes]
                reset_data["actions"] = np.zeros((1, len(dones_idxes), int(np.sum(actions_dim))), dtype=np.float32)
                reset_data["rewards"] = step_data["rewards"][:, dones_idxes]
                reset_data["is_first"] = np.zeros_like(reset_data["terminated"])
                rb.add(reset_data, dones_idxes)
                step_data["rewards"][:, dones_idxes] = 0.0
                step_data["terminated"][:, dones_idxes] = 0.0
                step_data["truncated"][:, dones_idxes] = 0.0
                step_data["is_first"][:, dones_idxes] = 1.0
                # player states are inference tensors (created under the
                # action-selection inference_mode); reset them under it too
                with torch.inference_mode():
                    player.init_states(dones_idxes)

        # training phase, replay-ratio driven.  On CUDA the whole gradient
        # step is hipGraph-captured after warmup (algo.hip_graphs, default
        # on): ~20k kernel launches collapse into one replay.  Metrics can't
        # cross the capture (host syncs), so every 16th gradient step runs
        # eager to feed the aggregator.
        if isinstance(rb, EnvIndependentReplayBuffer):
            rb_ready = any(len(b) >= cfg.algo.per_rank_sequence_length for b in rb.buffer)
        else:
            rb_ready = any(
                ep[next(iter(ep))].shape[0] >= cfg.algo.per_rank_sequence_length for ep in rb.buffer
            )
        if iter_num >= learning_starts and rb_ready:
            per_rank_gradient_steps = ratio((policy_step - prefill_steps) / world_size)
            if per_rank_gradient_steps > 0:
                with timer("Time/train_time"):
                    for _ in range(per_rank_gradient_steps):
                        if (
                            cumulative_per_rank_gradient_steps % cfg.algo.critic.per_rank_target_network_update_freq
                            == 0
                        ):
                            tau = (
                                1.0
                                if cumulative_per_rank_gradient_steps == 0
                                else cfg.algo.critic.tau
                            )
                            ops.ema_update_(
                                list(target_critic.parameters()), list(critic.parameters()), tau
                            )
                        sample = rb.sample_tensors(
                            cfg.algo.per_rank_batch_size,
                            sequence_length=cfg.algo.per_rank_sequence_length,
                            n_samples=1,
                            device=device,
                            from_numpy=cfg.buffer.from_numpy,
                        )
                        batch = {k: v[0].to(device) for k, v in sample.items()}
                        if graphed_step is None:
                            train(
                                runtime,
                                world_model,
                                actor,
                                critic,
                                target_critic,
                                world_optimizer,
                                actor_optimizer,
                                critic_optimizer,
                                batch,
                                aggregator,
                                cfg,
                                is_continuous,
                                actions_dim,
                                moments,
                            )
                        else:
                            graphed_step(batch)
                            # metrics come from the capture's static loss/
                            # grad buffers — device-side clones, no host
                            # sync, no eager metric steps (an eager step is
                            # ~14x a replay)
                            if aggregator and not MetricAggregator.disabled and graph_metrics:
                                _feed_graph_metrics(aggregator, graph_metrics)
                        cumulative_per_rank_gradient_steps += 1
                        # capture once the shapes/allocator have settled
                        if (
                            graphed_step is None
                            and want_graphs
                            and cumulative_per_rank_gradient_steps >= 3
                        ):
                            graphed_step, graph_metrics = _capture_train_step(
                                runtime, world_model, actor, critic, target_critic,
                                world_optimizer, actor_optimizer, critic_optimizer,
                                batch, cfg, is_continuous, actions_dim, moments,
                            )
                            if graphed_step is None:
                                want_graphs = False

        # logging
        if policy_step - last_log >= cfg.metric.log_every or iter_num == total_iters or cfg.dry_run:
            metrics = aggregator.compute()
            metrics["Params/replay_ratio"] = (
                cumulative_per_rank_gradient_steps * world_size / max(policy_step, 1)
            )
            times = timer.compute()
            if times.get("Time/train_time"):
                metrics["Time/sps_train"] = (
                    cumulative_per_rank_gradient_steps * world_size
                ) / times["Time/train_time"]
            if times.get("Time/env_interaction_time"):
                metrics["Time/sps_env_interaction"] = (
                    ((policy_step - last_log) / world_size) * cfg.env.action_repeat
                ) / times["Time/env_interaction_time"]
            runtime.log_dict(metrics, policy_step)
            aggregator.reset()
            timer.reset()
            last_log = policy_step

        # checkpoint
        if (
            cfg.checkpoint.every > 0
            and policy_step - last_checkpoint >= cfg.checkpoint.every
            or cfg.dry_run
            or (iter_num == total_iters and cfg.checkpoint.save_last)
        ):
            last_checkpoint = policy_step
            ckpt_state = {
                "world_model": world_model,
                "actor": actor,
                "critic": critic,
                "target_critic": target_critic,
                "world_optimizer": world_optimizer,
                "actor_optimizer": actor_optimizer,
                "critic_optimizer": critic_optimizer,
                "moments": moments,
                "ratio": ratio,
                "iter_num": iter_num + 1,
                "policy_step": policy_step,
                "batch_size": cfg.algo.per_rank_batch_size * world_size,
                "last_log": last_log,
                "last_checkpoint": last_checkpoint,
            }
            ckpt_path = os.path.join(log_dir, "checkpoint", f"ckpt_{policy_step}_{runtime.global_rank}.ckpt")
            runtime.call(
                "on_checkpoint_coupled",
                ckpt_path=ckpt_path,
                state=ckpt_state,
                replay_buffer=rb if cfg.buffer.get("checkpoint", False) else None,
            )

    envs.close()
    if runtime.is_global_zero and cfg.algo.run_test:
        reward = test(player, runtime, make_env(cfg, cfg.seed, 0), cfg, log_dir)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()
