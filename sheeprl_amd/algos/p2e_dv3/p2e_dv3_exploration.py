"""Plan2Explore DV3 — exploration phase.

Parity: sheeprl/algos/p2e_dv3/p2e_dv3_exploration.py — main :522, train :41:
1. DV3 world-model learning; 2. ensemble learning (MSE to the next posterior,
:205-231); 3. exploration behaviour: imagination driven by the exploration
actor, per-critic weighted advantages with intrinsic reward = ensemble
disagreement (variance, :270-285); 4. task behaviour trained exactly as DV3
so finetuning can start from it.  The player acts with the exploration actor.
"""

from __future__ import annotations

import os
from typing import Any, Dict, Sequence

import numpy as np
import torch
import torch.distributions as td
import torch.nn.functional as F

from sheeprl_amd import ops
from sheeprl_amd.algos.dreamer_v3.imagine import imagine_applicable, imagine_rollout
from sheeprl_amd.algos.dreamer_v3.loss import reconstruction_loss
from sheeprl_amd.algos.dreamer_v3.utils import Moments, compute_lambda_values, prepare_obs, test
from sheeprl_amd.algos.p2e_dv3.agent import build_agent
from sheeprl_amd.config import save_config
from sheeprl_amd.data import EnvIndependentReplayBuffer, SequentialReplayBuffer
from sheeprl_amd.distributions import BernoulliSafeMode, MSEDistribution, SymlogDistribution, TwoHotEncodingDistribution
from sheeprl_amd.envs import make_env, spaces, vectorize_env
from sheeprl_amd.optim import make_optimizer
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.logger import get_log_dir, get_logger
from sheeprl_amd.utils.metric import MetricAggregator
from sheeprl_amd.utils.registry import register_algorithm, register_evaluation
from sheeprl_amd.utils.timer import timer
from sheeprl_amd.utils.utils import Ratio
from contextlib import nullcontext as _nullcontext

AGGREGATOR_KEYS = {
    "Rewards/rew_avg",
    "Game/ep_len_avg",
    "Loss/world_model_loss",
    "Loss/policy_loss_task",
    "Loss/value_loss_task",
    "Loss/policy_loss_exploration",
    "Loss/ensemble_loss",
    "State/kl",
    "Rewards/intrinsic_intrinsic",
}
MODELS_TO_REGISTER = {"world_model", "ensembles", "actor_task", "critic_task", "target_critic_task", "actor_exploration"}


def train(
    runtime: Runtime,
    world_model,
    ensembles,
    actor_task,
    critic_task,
    target_critic_task,
    actor_exploration,
    critics_exploration: Dict[str, Dict[str, Any]],
    world_optimizer,
    ensemble_optimizer,
    actor_task_optimizer,
    critic_task_optimizer,
    actor_exploration_optimizer,
    critics_exploration_optimizers: Dict[str, Any],
    moments_task: Moments,
    data: Dict[str, torch.Tensor],
    aggregator,
    cfg: Any,
    is_continuous: bool,
    actions_dim: Sequence[int],
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
    batch_actions = torch.cat((torch.zeros_like(data["actions"][:1]), data["actions"][:-1]), dim=0)

    # ---- world model (same as DV3) ----
    recurrent_state = torch.zeros(1, batch_size, recurrent_state_size, device=device, dtype=dtype)
    recurrent_states = torch.empty(sequence_length, batch_size, recurrent_state_size, device=device, dtype=dtype)
    priors_logits = torch.empty(sequence_length, batch_size, stoch_state_size, device=device)
    posterior = torch.zeros(1, batch_size, stochastic_size, discrete_size, device=device, dtype=dtype)
    posteriors = torch.empty(sequence_length, batch_size, stochastic_size, discrete_size, device=device, dtype=dtype)
    posteriors_logits = torch.empty(sequence_length, batch_size, stoch_state_size, device=device)
    embedded_obs = world_model.encoder(batch_obs)
    _ir, _ip = world_model.rssm.get_initial_states((1, batch_size))
    initial_states = (_ir.contiguous(), _ip.contiguous())
    for i in range(sequence_length):
        recurrent_state, posterior, posterior_logits = world_model.rssm.dynamic_posterior(
            posterior, recurrent_state, batch_actions[i : i + 1], embedded_obs[i : i + 1],
            data["is_first"][i : i + 1], initial_states=initial_states,
        )
        recurrent_states[i] = recurrent_state
        posteriors[i] = posterior
        posteriors_logits[i] = posterior_logits
    priors_logits = world_model.rssm.transition_logits(recurrent_states)
    latent_states = torch.cat((posteriors.view(*posteriors.shape[:-2], -1), recurrent_states), -1)

    reconstructed_obs = world_model.observation_model(latent_states)
    po = {
        k: MSEDistribution(reconstructed_obs[k].float(), dims=len(reconstructed_obs[k].shape[2:]))
        for k in cfg.algo.cnn_keys.decoder
    }
    po.update(
        {
            k: SymlogDistribution(reconstructed_obs[k].float(), dims=len(reconstructed_obs[k].shape[2:]))
            for k in cfg.algo.mlp_keys.decoder
        }
    )
    pr = TwoHotEncodingDistribution(world_model.reward_model(latent_states).float(), dims=1)
    pc = td.Independent(BernoulliSafeMode(logits=world_model.continue_model(latent_states).float()), 1)
    continues_targets = 1 - data["terminated"]
    priors_logits = priors_logits.view(*priors_logits.shape[:-1], stochastic_size, discrete_size)
    posteriors_logits = posteriors_logits.view(*posteriors_logits.shape[:-1], stochastic_size, discrete_size)

    world_optimizer.zero_grad(set_to_none=True)
    rec_loss, kl, state_loss, reward_loss, observation_loss, continue_loss = reconstruction_loss(
        po, {k: v.float() for k, v in batch_obs.items()}, pr, data["rewards"],
        priors_logits, posteriors_logits,
        cfg.algo.world_model.kl_dynamic, cfg.algo.world_model.kl_representation,
        cfg.algo.world_model.kl_free_nats, cfg.algo.world_model.kl_regularizer,
        pc, continues_targets, cfg.algo.world_model.continue_scale_factor,
    )
    runtime.backward(rec_loss)
    if cfg.algo.world_model.clip_gradients and cfg.algo.world_model.clip_gradients > 0:
        runtime.clip_gradients(world_model, world_optimizer, cfg.algo.world_model.clip_gradients)
    world_optimizer.step()

    # ---- ensemble learning (p2e_dv3_exploration.py:205-231) ----
    ensemble_optimizer.zero_grad(set_to_none=True)
    ens_loss = 0.0
    ens_input = torch.cat(
        (posteriors.view(*posteriors.shape[:-2], -1).detach(), recurrent_states.detach(), data["actions"].detach().to(dtype)),
        -1,
    )
    target_next = posteriors.view(sequence_length, batch_size, -1).detach()[1:].float()
    for ens in ensembles:
        out = ens(ens_input)[:-1].float()
        ens_loss = ens_loss - MSEDistribution(out, 1).log_prob(target_next).mean()
    runtime.backward(ens_loss)
    if cfg.algo.ensembles.clip_gradients and cfg.algo.ensembles.clip_gradients > 0:
        runtime.clip_gradients(ensembles, ensemble_optimizer, cfg.algo.ensembles.clip_gradients)
    ensemble_optimizer.step()

    # ---- exploration behaviour ----
    horizon = cfg.algo.horizon
    flat = batch_size * sequence_length
    # DV3 fast path (launch-lean no-grad rollout + fused REINFORCE/critic
    # losses): valid because nothing backpropagates through the imagined
    # rollout for a discrete single-head actor (every consumer detaches)
    use_fast = (
        cfg.algo.get("fused_imagination", True)
        and not is_continuous
        and len(actions_dim) == 1
        and device.type == "cuda"
        and imagine_applicable(world_model.rssm, actor_exploration)
        and imagine_applicable(world_model.rssm, actor_task)
    )
    if use_fast:
        imagined_trajectories, imagined_actions = imagine_rollout(
            world_model.rssm,
            actor_exploration,
            posteriors.detach().reshape(flat, stoch_state_size).to(dtype),
            recurrent_states.detach().reshape(flat, recurrent_state_size).to(dtype),
            horizon,
        )
    else:
        imagined_prior = posteriors.detach().reshape(1, -1, stoch_state_size)
        recurrent_state = recurrent_states.detach().reshape(1, -1, recurrent_state_size)
        imagined_latent_state = torch.cat((imagined_prior, recurrent_state), -1)
        imagined_trajectories = torch.empty(horizon + 1, flat, stoch_state_size + recurrent_state_size,
                                            device=device, dtype=dtype)
        imagined_trajectories[0] = imagined_latent_state
        imagined_actions = torch.empty(horizon + 1, flat, data["actions"].shape[-1], device=device, dtype=dtype)
        actions = torch.cat(actor_exploration(imagined_latent_state.detach())[0], dim=-1).to(dtype)
        imagined_actions[0] = actions
        for i in range(1, horizon + 1):
            imagined_prior, recurrent_state = world_model.rssm.imagination(imagined_prior, recurrent_state, actions)
            imagined_prior = imagined_prior.view(1, -1, stoch_state_size).to(dtype)
            imagined_latent_state = torch.cat((imagined_prior, recurrent_state), -1)
            imagined_trajectories[i] = imagined_latent_state
            actions = torch.cat(actor_exploration(imagined_latent_state.detach())[0], dim=-1).to(dtype)
            imagined_actions[i] = actions

    weights_sum = sum(c["weight"] for c in critics_exploration.values())
    advantages = []
    for name, critic in critics_exploration.items():
        if use_fast:
            with torch.no_grad():
                predicted_values = ops.twohot_mean(critic["module"](imagined_trajectories))
                continues = (world_model.continue_model(imagined_trajectories) > 0).float()
                true_continue = (1 - data["terminated"]).flatten().reshape(1, -1, 1)
                continues = torch.cat((true_continue, continues[1:]))
        else:
            predicted_values = TwoHotEncodingDistribution(critic["module"](imagined_trajectories).float(), dims=1).mean
            continues = td.Independent(
                BernoulliSafeMode(logits=world_model.continue_model(imagined_trajectories).float()), 1
            ).mode
            true_continue = (1 - data["terminated"]).flatten().reshape(1, -1, 1)
            continues = torch.cat((true_continue, continues[1:]))
        if critic["reward_type"] == "intrinsic":
            with torch.no_grad():
                next_state_embedding = torch.stack(
                    [
                        ens(torch.cat((imagined_trajectories.detach(), imagined_actions.detach()), -1)).float()
                        for ens in ensembles
                    ]
                )
            reward = next_state_embedding.var(0).mean(-1, keepdim=True) * cfg.algo.intrinsic_reward_multiplier
            if aggregator and not MetricAggregator.disabled:
                aggregator.update(f"Rewards/intrinsic_{name}", reward.detach().mean())
        elif use_fast:
            with torch.no_grad():
                reward = ops.twohot_mean(world_model.reward_model(imagined_trajectories))
        else:
            reward = TwoHotEncodingDistribution(world_model.reward_model(imagined_trajectories).float(), dims=1).mean
        with torch.no_grad() if use_fast else _nullcontext():
            lambda_values = compute_lambda_values(
                reward[1:], predicted_values[1:], continues[1:] * cfg.algo.gamma, lmbda=cfg.algo.lmbda
            )
        critic["lambda_values"] = lambda_values.detach()
        critic["continues"] = continues.detach()
        baseline = predicted_values[:-1]
        offset, invscale = critic["moments"](lambda_values, runtime)
        if use_fast:
            # offsets cancel: ((λ-off) - (v-off)) / s = (λ - v) / s
            with torch.no_grad():
                advantages.append((lambda_values - baseline) / invscale * (critic["weight"] / weights_sum))
        else:
            advantages.append(((lambda_values - offset) / invscale - (baseline - offset) / invscale)
                              * critic["weight"] / weights_sum)

    advantage = sum(advantages)
    with torch.no_grad():
        continues = critics_exploration[next(iter(critics_exploration))]["continues"]
        discount = torch.cumprod(continues * cfg.algo.gamma, dim=0) / cfg.algo.gamma

    actor_exploration_optimizer.zero_grad(set_to_none=True)
    policies = actor_exploration(imagined_trajectories.detach())[1]
    if use_fast:
        policy_loss_expl = ops.reinforce_loss(
            policies[0].logits, imagined_actions, advantage, discount[:-1], cfg.algo.actor.ent_coef
        )
    else:
        if is_continuous:
            objective = advantage
        else:
            objective = (
                torch.stack(
                    [
                        p.log_prob(a.detach().float()).unsqueeze(-1)[:-1]
                        for p, a in zip(policies, torch.split(imagined_actions, list(actions_dim), -1))
                    ],
                    -1,
                ).sum(-1)
                * advantage.detach()
            )
        try:
            entropy = cfg.algo.actor.ent_coef * torch.stack([p.entropy() for p in policies], -1).sum(-1)
        except NotImplementedError:
            entropy = torch.zeros_like(objective)
        policy_loss_expl = -torch.mean(discount[:-1].detach() * (objective + entropy.unsqueeze(-1)[:-1]))
    runtime.backward(policy_loss_expl)
    if cfg.algo.actor.clip_gradients and cfg.algo.actor.clip_gradients > 0:
        runtime.clip_gradients(actor_exploration, actor_exploration_optimizer, cfg.algo.actor.clip_gradients)
    actor_exploration_optimizer.step()

    # exploration critics
    for name, critic in critics_exploration.items():
        opt = critics_exploration_optimizers[name]
        opt.zero_grad(set_to_none=True)
        if use_fast:
            qv_logits = critic["module"](imagined_trajectories.detach()[:-1]).float()
            with torch.no_grad():
                predicted_target = ops.twohot_mean(critic["target_module"](imagined_trajectories.detach()[:-1]))
            value_loss = ops.critic_twohot_loss(
                qv_logits, critic["lambda_values"], predicted_target, discount[:-1]
            )
        else:
            qv = TwoHotEncodingDistribution(critic["module"](imagined_trajectories.detach()[:-1]).float(), dims=1)
            predicted_target = TwoHotEncodingDistribution(
                critic["target_module"](imagined_trajectories.detach()[:-1]).float(), dims=1
            ).mean
            value_loss = -qv.log_prob(critic["lambda_values"].detach()) - qv.log_prob(predicted_target.detach())
            value_loss = torch.mean(value_loss * discount[:-1].squeeze(-1))
        runtime.backward(value_loss)
        if cfg.algo.critic.clip_gradients and cfg.algo.critic.clip_gradients > 0:
            runtime.clip_gradients(critic["module"], opt, cfg.algo.critic.clip_gradients)
        opt.step()
        ops.ema_update_(
            list(critic["target_module"].parameters()), list(critic["module"].parameters()), cfg.algo.critic.tau
        )

    # ---- task behaviour (identical math to DV3, imagination by the task actor) ----
    if use_fast:
        imagined_trajectories, imagined_actions = imagine_rollout(
            world_model.rssm,
            actor_task,
            posteriors.detach().reshape(flat, stoch_state_size).to(dtype),
            recurrent_states.detach().reshape(flat, recurrent_state_size).to(dtype),
            horizon,
        )
        with torch.no_grad():
            predicted_values = ops.twohot_mean(critic_task(imagined_trajectories))
            predicted_rewards = ops.twohot_mean(world_model.reward_model(imagined_trajectories))
            continues = (world_model.continue_model(imagined_trajectories) > 0).float()
            true_continue = (1 - data["terminated"]).flatten().reshape(1, -1, 1)
            continues = torch.cat((true_continue, continues[1:]))
            lambda_values = compute_lambda_values(
                predicted_rewards[1:], predicted_values[1:], continues[1:] * cfg.algo.gamma, lmbda=cfg.algo.lmbda
            )
            discount = torch.cumprod(continues * cfg.algo.gamma, dim=0) / cfg.algo.gamma

        actor_task_optimizer.zero_grad(set_to_none=True)
        policies = actor_task(imagined_trajectories.detach())[1]
        offset, invscale = moments_task(lambda_values, runtime)
        with torch.no_grad():
            advantage = (lambda_values - predicted_values[:-1]) / invscale
        policy_loss_task = ops.reinforce_loss(
            policies[0].logits, imagined_actions, advantage, discount[:-1], cfg.algo.actor.ent_coef
        )
    else:
        imagined_prior = posteriors.detach().reshape(1, -1, stoch_state_size)
        recurrent_state = recurrent_states.detach().reshape(1, -1, recurrent_state_size)
        imagined_latent_state = torch.cat((imagined_prior, recurrent_state), -1)
        imagined_trajectories = torch.empty(horizon + 1, flat, stoch_state_size + recurrent_state_size,
                                            device=device, dtype=dtype)
        imagined_trajectories[0] = imagined_latent_state
        imagined_actions = torch.empty(horizon + 1, flat, data["actions"].shape[-1], device=device, dtype=dtype)
        actions = torch.cat(actor_task(imagined_latent_state.detach())[0], dim=-1).to(dtype)
        imagined_actions[0] = actions
        for i in range(1, horizon + 1):
            imagined_prior, recurrent_state = world_model.rssm.imagination(imagined_prior, recurrent_state, actions)
            imagined_prior = imagined_prior.view(1, -1, stoch_state_size).to(dtype)
            imagined_latent_state = torch.cat((imagined_prior, recurrent_state), -1)
            imagined_trajectories[i] = imagined_latent_state
            actions = torch.cat(actor_task(imagined_latent_state.detach())[0], dim=-1).to(dtype)
            imagined_actions[i] = actions

        predicted_values = TwoHotEncodingDistribution(critic_task(imagined_trajectories).float(), dims=1).mean
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

        actor_task_optimizer.zero_grad(set_to_none=True)
        policies = actor_task(imagined_trajectories.detach())[1]
        baseline = predicted_values[:-1]
        offset, invscale = moments_task(lambda_values, runtime)
        advantage = (lambda_values - offset) / invscale - (baseline - offset) / invscale
        if is_continuous:
            objective = advantage
        else:
            objective = (
                torch.stack(
                    [
                        p.log_prob(a.detach().float()).unsqueeze(-1)[:-1]
                        for p, a in zip(policies, torch.split(imagined_actions, list(actions_dim), -1))
                    ],
                    -1,
                ).sum(-1)
                * advantage.detach()
            )
        try:
            entropy = cfg.algo.actor.ent_coef * torch.stack([p.entropy() for p in policies], -1).sum(-1)
        except NotImplementedError:
            entropy = torch.zeros_like(objective)
        policy_loss_task = -torch.mean(discount[:-1].detach() * (objective + entropy.unsqueeze(-1)[:-1]))
    runtime.backward(policy_loss_task)
    if cfg.algo.actor.clip_gradients and cfg.algo.actor.clip_gradients > 0:
        runtime.clip_gradients(actor_task, actor_task_optimizer, cfg.algo.actor.clip_gradients)
    actor_task_optimizer.step()

    critic_task_optimizer.zero_grad(set_to_none=True)
    if use_fast:
        qv_logits = critic_task(imagined_trajectories.detach()[:-1]).float()
        with torch.no_grad():
            predicted_target_values = ops.twohot_mean(target_critic_task(imagined_trajectories.detach()[:-1]))
        value_loss_task = ops.critic_twohot_loss(
            qv_logits, lambda_values, predicted_target_values, discount[:-1]
        )
    else:
        qv = TwoHotEncodingDistribution(critic_task(imagined_trajectories.detach()[:-1]).float(), dims=1)
        predicted_target_values = TwoHotEncodingDistribution(
            target_critic_task(imagined_trajectories.detach()[:-1]).float(), dims=1
        ).mean
        value_loss_task = -qv.log_prob(lambda_values.detach()) - qv.log_prob(predicted_target_values.detach())
        value_loss_task = torch.mean(value_loss_task * discount[:-1].squeeze(-1))
    runtime.backward(value_loss_task)
    if cfg.algo.critic.clip_gradients and cfg.algo.critic.clip_gradients > 0:
        runtime.clip_gradients(critic_task, critic_task_optimizer, cfg.algo.critic.clip_gradients)
    critic_task_optimizer.step()
    ops.ema_update_(list(target_critic_task.parameters()), list(critic_task.parameters()), cfg.algo.critic.tau)

    if aggregator and not MetricAggregator.disabled:
        aggregator.update("Loss/world_model_loss", rec_loss.detach())
        aggregator.update("State/kl", kl.mean().detach())
        aggregator.update("Loss/ensemble_loss", ens_loss.detach())
        aggregator.update("Loss/policy_loss_exploration", policy_loss_expl.detach())
        aggregator.update("Loss/policy_loss_task", policy_loss_task.detach())
        aggregator.update("Loss/value_loss_task", value_loss_task.detach())


def _capture_train_step(train_args):
    """hipGraph-capture the P2E-DV3 gradient step (unconditional in-place
    EMA/moments updates make the whole step replayable); None on failure."""
    from sheeprl_amd.parallel.graphs import CUDAGraphStep
    from sheeprl_amd.utils.metric import MetricAggregator

    runtime = train_args[0]
    pre, post = train_args[:15], train_args[16:]

    def train_fn(batch):
        was_disabled = MetricAggregator.disabled
        MetricAggregator.disabled = True
        try:
            train(*pre, batch, None, *post[1:])
        finally:
            MetricAggregator.disabled = was_disabled

    try:
        step = CUDAGraphStep(train_fn, train_args[15], warmup=2)
        runtime.print("[p2e_dv3] gradient step captured in a hipGraph")
        return step
    except Exception as e:  # noqa: BLE001
        runtime.print(f"[p2e_dv3] hipGraph capture failed ({e}); eager training")
        return None


@register_algorithm(name="p2e_dv3_exploration")
def main(runtime: Runtime, cfg: Any) -> None:
    device = runtime.device
    log_dir = get_log_dir(runtime, cfg.root_dir, cfg.run_name)
    logger = get_logger(runtime, cfg, log_dir)
    runtime.logger = logger
    if runtime.is_global_zero:
        save_config(cfg, os.path.join(log_dir, "config.yaml"))

    cfg.algo.player.actor_type = "exploration"
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

    (
        world_model, ensembles, actor_task, critic_task, target_critic_task,
        actor_exploration, critics_exploration, player,
    ) = build_agent(
        runtime, actions_dim, is_continuous, cfg, obs_space,
        state.get("world_model"), state.get("ensembles"), state.get("actor_task"),
        state.get("critic_task"), state.get("target_critic_task"),
        state.get("actor_exploration"), state.get("critics_exploration"),
    )

    world_optimizer = make_optimizer(world_model.parameters(), cfg.algo.world_model.optimizer)
    ensemble_optimizer = make_optimizer(ensembles.parameters(), cfg.algo.ensembles.optimizer)
    actor_task_optimizer = make_optimizer(actor_task.parameters(), cfg.algo.actor.optimizer)
    critic_task_optimizer = make_optimizer(critic_task.parameters(), cfg.algo.critic.optimizer)
    actor_expl_optimizer = make_optimizer(actor_exploration.parameters(), cfg.algo.actor.optimizer)
    critics_expl_optimizers = {
        name: make_optimizer(c["module"].parameters(), cfg.algo.critic.optimizer)
        for name, c in critics_exploration.items()
    }
    moments_task = Moments(
        cfg.algo.actor.moments.decay, cfg.algo.actor.moments.max,
        cfg.algo.actor.moments.percentile.low, cfg.algo.actor.moments.percentile.high,
    ).to(device)

    aggregator = MetricAggregator({k: "mean" for k in AGGREGATOR_KEYS})
    rb = EnvIndependentReplayBuffer(
        max(int(cfg.buffer.size), 1), n_envs=cfg.env.num_envs, obs_keys=obs_keys,
        buffer_cls=SequentialReplayBuffer,
    )

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
    policy_step = int(state.get("policy_step", 0))
    last_log = 0
    graphed_step = None
    _n_train_calls = 0
    # round-2: capture is ON by default — the round-1 segfault was the
    # behaviour tensors (lambda_values/continues) stored with live autograd
    # graphs across steps, keeping stale AccumulateGrad nodes alive into the
    # capture stream; they are stored detached now
    want_graphs = (
        (runtime.device.type == "cuda" or os.environ.get("SHEEPRL_AMD_FORCE_GRAPHS") == "1")
        and cfg.algo.get("hip_graphs", True)
        and os.environ.get("SHEEPRL_AMD_NO_GRAPHS", "0") != "1"
    )
    last_checkpoint = 0
    ratio = Ratio(cfg.algo.replay_ratio, pretrain_steps=cfg.algo.per_rank_pretrain_steps)

    step_data: Dict[str, np.ndarray] = {}
    obs, _ = envs.reset(seed=cfg.seed + runtime.global_rank * num_envs)
    for k in obs_keys:
        step_data[k] = np.asarray(obs[k])[np.newaxis]
    step_data["rewards"] = np.zeros((1, num_envs, 1), dtype=np.float32)
    step_data["truncated"] = np.zeros((1, num_envs, 1), dtype=np.float32)
    step_data["terminated"] = np.zeros((1, num_envs, 1), dtype=np.float32)
    step_data["is_first"] = np.ones_like(step_data["terminated"])
    player.init_states()

    for iter_num in range(1, total_iters + 1):
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
                step_data["actions"] = np.asarray(actions, np.float32).reshape(1, num_envs, -1)
                rb.add(step_data)
                next_obs, rewards, terminated, truncated, infos = envs.step(real_actions)
                dones = np.logical_or(terminated, truncated)

            step_data["is_first"] = np.zeros_like(step_data["terminated"])
            for ep in infos.get("episode", []):
                if ep is not None:
                    aggregator.update("Rewards/rew_avg", float(ep["r"][0]))
                    aggregator.update("Game/ep_len_avg", float(ep["l"][0]))
            real_next_obs = {k: np.asarray(v).copy() for k, v in next_obs.items()}
            for idx, fo in enumerate(infos.get("final_observation", [])):
                if fo is not None:
                    for k in obs_keys:
                        real_next_obs[k][idx] = fo[k]
            for k in obs_keys:
                step_data[k] = np.asarray(next_obs[k])[np.newaxis]
            obs = next_obs
            step_data["rewards"] = np.asarray(rewards, np.float32).reshape(1, num_envs, 1)
            step_data["terminated"] = np.asarray(terminated, np.float32).reshape(1, num_envs, 1)
            step_data["truncated"] = np.asarray(truncated, np.float32).reshape(1, num_envs, 1)
            dones_idxes = np.nonzero(dones)[0].tolist()
            if dones_idxes:
                reset_data = {}
                for k in obs_keys:
                    reset_data[k] = real_next_obs[k][dones_idxes][np.newaxis]
                reset_data["terminated"] = step_data["terminated"][:, dones_idxes]
                reset_data["truncated"] = step_data["truncated"][:, dones_idxes]
                reset_data["actions"] = np.zeros((1, len(dones_idxes), int(np.sum(actions_dim))), np.float32)
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

        rb_ready = any(len(b) >= cfg.algo.per_rank_sequence_length for b in rb.buffer)
        if iter_num >= learning_starts and rb_ready:
            n_steps = ratio((policy_step - prefill_steps) / world_size)
            if n_steps > 0:
                with timer("Time/train_time"):
                    for _ in range(n_steps):
                        sample = rb.sample_tensors(
                            cfg.algo.per_rank_batch_size,
                            sequence_length=cfg.algo.per_rank_sequence_length,
                            n_samples=1,
                            device=device,
                        )
                        batch = {k: v[0].to(device) for k, v in sample.items()}
                        _n_train_calls += 1
                        if graphed_step is not None and _n_train_calls % 64 != 0:  # every 64th step runs eager to feed metrics (~14x a replay; DV3 feeds metrics from the capture's static buffers instead)
                            graphed_step(batch)
                        else:
                            train(
                                runtime, world_model, ensembles, actor_task, critic_task, target_critic_task,
                                actor_exploration, critics_exploration, world_optimizer, ensemble_optimizer,
                                actor_task_optimizer, critic_task_optimizer, actor_expl_optimizer,
                                critics_expl_optimizers, moments_task, batch, aggregator, cfg,
                                is_continuous, actions_dim,
                            )
                        if graphed_step is None and want_graphs and _n_train_calls >= 3:
                            graphed_step = _capture_train_step((
                                runtime, world_model, ensembles, actor_task, critic_task, target_critic_task,
                                actor_exploration, critics_exploration, world_optimizer, ensemble_optimizer,
                                actor_task_optimizer, critic_task_optimizer, actor_expl_optimizer,
                                critics_expl_optimizers, moments_task, batch, aggregator, cfg,
                                is_continuous, actions_dim,
                            ))
                            if graphed_step is None:
                                want_graphs = False

        if policy_step - last_log >= cfg.metric.log_every or iter_num == total_iters or cfg.dry_run:
            runtime.log_dict(aggregator.compute(), policy_step)
            aggregator.reset()
            timer.reset()
            last_log = policy_step

        if (
            cfg.checkpoint.every > 0
            and policy_step - last_checkpoint >= cfg.checkpoint.every
            or cfg.dry_run
            or (iter_num == total_iters and cfg.checkpoint.save_last)
        ):
            last_checkpoint = policy_step
            ckpt_path = os.path.join(log_dir, "checkpoint", f"ckpt_{policy_step}_{runtime.global_rank}.ckpt")
            runtime.call(
                "on_checkpoint_coupled",
                ckpt_path=ckpt_path,
                state={
                    "world_model": world_model,
                    "ensembles": ensembles,
                    "actor_task": actor_task,
                    "critic_task": critic_task,
                    "target_critic_task": target_critic_task,
                    "actor_exploration": actor_exploration,
                    "critics_exploration": {
                        name: {
                            "module": c["module"].state_dict(),
                            "target": c["target_module"].state_dict(),
                            "moments": c["moments"].state_dict(),
                        }
                        for name, c in critics_exploration.items()
                    },
                    "moments_task": moments_task,
                    "ratio": ratio,
                    "policy_step": policy_step,
                    "batch_size": cfg.algo.per_rank_batch_size * world_size,
                },
                replay_buffer=rb if cfg.buffer.get("checkpoint", False) else None,
            )

    envs.close()
    if runtime.is_global_zero and cfg.algo.run_test:
        player.actor = actor_task  # test with the task policy (reference :1032)
        player.actor_type = "task"
        reward = test(player, runtime, make_env(cfg, cfg.seed, 0), cfg, log_dir)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()


@register_evaluation(algorithms=["p2e_dv3_exploration"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    env_fn = make_env(cfg, cfg.seed, 0)
    env = env_fn()
    obs_space, action_space = env.observation_space, env.action_space
    env.close()
    is_continuous = isinstance(action_space, spaces.Box)
    is_multidiscrete = isinstance(action_space, spaces.MultiDiscrete)
    actions_dim = tuple(
        action_space.shape if is_continuous else (action_space.nvec.tolist() if is_multidiscrete else [action_space.n])
    )
    cfg.algo.player.actor_type = "task"
    out = build_agent(
        runtime, actions_dim, is_continuous, cfg, obs_space,
        state["world_model"], state.get("ensembles"), state.get("actor_task"),
        state.get("critic_task"), state.get("target_critic_task"), state.get("actor_exploration"),
        state.get("critics_exploration"),
    )
    player = out[-1]
    reward = test(player, runtime, env_fn, cfg)
    runtime.print(f"Test/cumulative_reward: {reward}")
    return reward
