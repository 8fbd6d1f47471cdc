"""Dreamer-V2 training loop.

Parity: sheeprl/algos/dreamer_v2/dreamer_v2.py — train :41 (two-phase update
documented :55-87), main :390; loss.py:9 (alpha-balanced KL :60-85);
compute_lambda_values variant dreamer_v2/utils.py:85-102; hard target-critic
copy every ``per_rank_target_network_update_freq`` gradient steps; buffer
selectable sequential/episode (dreamer_v2.py:496-517).
"""

from __future__ import annotations

import os
from typing import Any, Dict, Sequence

import numpy as np
import torch
import torch.distributions as td
import torch.nn.functional as F

from sheeprl_amd import ops
from sheeprl_amd.algos.dreamer_v2.agent import build_agent
from sheeprl_amd.algos.dreamer_v3.loss import categorical_kl
from sheeprl_amd.algos.dreamer_v3.utils import prepare_obs, test
from sheeprl_amd.config import save_config
from sheeprl_amd.data import EnvIndependentReplayBuffer, EpisodeBuffer, SequentialReplayBuffer
from sheeprl_amd.envs import make_env, spaces, vectorize_env
from sheeprl_amd.optim import make_optimizer
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.logger import get_log_dir, get_logger
from sheeprl_amd.utils.metric import MetricAggregator
from sheeprl_amd.utils.registry import register_algorithm, register_evaluation
from sheeprl_amd.utils.timer import timer
from sheeprl_amd.utils.utils import Ratio

AGGREGATOR_KEYS = {
    "Rewards/rew_avg",
    "Game/ep_len_avg",
    "Loss/world_model_loss",
    "Loss/value_loss",
    "Loss/policy_loss",
    "Loss/observation_loss",
    "Loss/reward_loss",
    "Loss/state_loss",
    "Loss/continue_loss",
    "State/kl",
}
MODELS_TO_REGISTER = {"world_model", "actor", "critic", "target_critic"}



def _unit_scale(x):
    """Scale=1 as a device tensor: td.Normal(x, 1) materializes the python
    scalar with a pageable H2D copy, which is illegal inside hipGraph capture."""
    return torch.ones((), device=x.device, dtype=x.dtype)


class _FastUnitNormal:
    """Independent(Normal(loc, 1), n) restricted to the members the DV2 loss
    path touches.  td.Normal.log_prob builds several full-size temporaries
    for the scale broadcast (log sigma, division, var) — measured 0.55 s per
    CPU train step over image tensors; unit scale needs one sub, one square:
    log N(x; mu, 1) = -0.5 (x-mu)^2 - 0.5 log(2 pi)."""

    _LOG_SQRT_2PI = 0.9189385332046727

    def __init__(self, loc, event_dims: int):
        self.loc = loc
        self._n = event_dims

    @property
    def mean(self):
        return self.loc

    @property
    def mode(self):
        return self.loc

    def log_prob(self, x):
        lp = -0.5 * (x - self.loc).pow(2) - self._LOG_SQRT_2PI
        return lp.sum(tuple(range(-self._n, 0))) if self._n else lp

def dv2_reconstruction_loss(
    po, observations, pr, rewards, priors_logits, posteriors_logits,
    kl_balancing_alpha=0.8, kl_free_nats=1.0, kl_free_avg=True, kl_regularizer=1.0,
    pc=None, continue_targets=None, discount_scale_factor=1.0,
):
    """Parity: dreamer_v2/loss.py:9-85 (alpha-balanced KL with free nats)."""
    observation_loss = -sum(po[k].log_prob(observations[k]).mean() for k in po.keys())
    reward_loss = -pr.log_prob(rewards).mean()
    lhs = kl = categorical_kl(posteriors_logits.detach(), priors_logits)
    rhs = categorical_kl(posteriors_logits, priors_logits.detach())
    if kl_free_avg:
        lhs, rhs = lhs.mean(), rhs.mean()
        free = torch.full_like(lhs, kl_free_nats)
        loss_lhs = torch.maximum(lhs, free)
        loss_rhs = torch.maximum(rhs, free)
    else:
        loss_lhs = torch.clamp(lhs, min=kl_free_nats).mean()
        loss_rhs = torch.clamp(rhs, min=kl_free_nats).mean()
    kl_loss = kl_balancing_alpha * loss_lhs + (1 - kl_balancing_alpha) * loss_rhs
    if pc is not None and continue_targets is not None:
        continue_loss = discount_scale_factor * -pc.log_prob(continue_targets).mean()
    else:
        continue_loss = torch.zeros_like(reward_loss)
    rec_loss = kl_regularizer * kl_loss + observation_loss + reward_loss + continue_loss
    return rec_loss, kl, kl_loss, reward_loss, observation_loss, continue_loss


def compute_lambda_values(rewards, values, continues, bootstrap=None, horizon: int = 15, lmbda: float = 0.95):
    """DV2 λ-values (parity: dreamer_v2/utils.py:85-102):
    L_t = r_t + c_t((1-λ) v_{t+1} + λ L_{t+1}), L_H = bootstrap."""
    if bootstrap is None:
        bootstrap = torch.zeros_like(values[-1:])
    next_values = torch.cat((values[1:], bootstrap), dim=0)
    return ops.lambda_values(rewards, next_values, continues, lmbda)


def train(
    runtime: Runtime,
    world_model,
    actor,
    critic,
    target_critic,
    world_optimizer,
    actor_optimizer,
    critic_optimizer,
    data: Dict[str, torch.Tensor],
    aggregator,
    cfg: Any,
    is_continuous: bool,
    actions_dim: Sequence[int],
    cumulative_step: int,
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

    # dynamic learning
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

    decoded = world_model.observation_model(latent_states)
    po = {k: _FastUnitNormal(v.float(), len(v.shape[2:])) for k, v in decoded.items()}
    _rm_out = world_model.reward_model(latent_states).float()

    pr = _FastUnitNormal(_rm_out, 1)
    if cfg.algo.world_model.use_continues and world_model.continue_model:
        pc = td.Independent(td.Bernoulli(logits=world_model.continue_model(latent_states).float()), 1)
        continues_targets = (1 - data["terminated"]) * cfg.algo.gamma
    else:
        pc = continues_targets = None

    priors_logits = priors_logits.view(*priors_logits.shape[:-1], stochastic_size, discrete_size)
    posteriors_logits = posteriors_logits.view(*posteriors_logits.shape[:-1], stochastic_size, discrete_size)

    world_optimizer.zero_grad(set_to_none=True)
    rec_loss, kl, state_loss, reward_loss, observation_loss, continue_loss = dv2_reconstruction_loss(
        po, {k: v.float() for k, v in batch_obs.items()}, pr, data["rewards"],
        priors_logits, posteriors_logits,
        cfg.algo.world_model.kl_balancing_alpha, cfg.algo.world_model.kl_free_nats,
        cfg.algo.world_model.get("kl_free_avg", True), cfg.algo.world_model.kl_regularizer,
        pc, continues_targets, cfg.algo.world_model.discount_scale_factor,
    )
    runtime.backward(rec_loss)
    if cfg.algo.world_model.clip_gradients and cfg.algo.world_model.clip_gradients > 0:
        runtime.clip_gradients(world_model, world_optimizer, cfg.algo.world_model.clip_gradients)
    world_optimizer.step()

    # behaviour learning (imagination with the TARGET critic, DV2 style)
    horizon = cfg.algo.horizon
    imagined_prior = posteriors.detach().reshape(1, -1, stoch_state_size)
    recurrent_state = recurrent_states.detach().reshape(1, -1, recurrent_state_size)
    imagined_latent_state = torch.cat((imagined_prior, recurrent_state), -1)
    flat = batch_size * sequence_length
    imagined_trajectories = torch.empty(horizon + 1, flat, stoch_state_size + recurrent_state_size,
                                        device=device, dtype=dtype)
    imagined_trajectories[0] = imagined_latent_state
    imagined_actions = torch.empty(horizon + 1, flat, data["actions"].shape[-1], device=device, dtype=dtype)
    imagined_actions[0] = 0.0
    for i in range(1, horizon + 1):
        actions = torch.cat(actor(imagined_latent_state.detach())[0], dim=-1).to(dtype)
        imagined_actions[i] = actions
        imagined_prior, recurrent_state = world_model.rssm.imagination(imagined_prior, recurrent_state, actions)
        imagined_prior = imagined_prior.view(1, -1, stoch_state_size).to(dtype)
        imagined_latent_state = torch.cat((imagined_prior, recurrent_state), -1)
        imagined_trajectories[i] = imagined_latent_state

    predicted_target_values = target_critic(imagined_trajectories).float()
    predicted_rewards = world_model.reward_model(imagined_trajectories).float()
    if cfg.algo.world_model.use_continues and world_model.continue_model:
        continues = torch.sigmoid(world_model.continue_model(imagined_trajectories).float())
        true_continue = (1 - data["terminated"]).reshape(1, -1, 1) * cfg.algo.gamma
        continues = torch.cat((true_continue, continues[1:]))
    else:
        continues = torch.ones_like(predicted_rewards.detach()) * cfg.algo.gamma

    lambda_values = compute_lambda_values(
        predicted_rewards[:-1], predicted_target_values[:-1], continues[:-1],
        bootstrap=predicted_target_values[-1:], horizon=horizon, lmbda=cfg.algo.lmbda,
    )
    with torch.no_grad():
        discount = torch.cumprod(torch.cat((torch.ones_like(continues[:1]), continues[:-1]), 0), 0)

    # actor: mixed reinforce + dynamics objective (dreamer_v2.py:291-326)
    actor_optimizer.zero_grad(set_to_none=True)
    policies = actor(imagined_trajectories[:-2].detach())[1]
    dynamics = lambda_values[1:]
    advantage = (lambda_values[1:] - predicted_target_values[:-2]).detach()
    reinforce = (
        torch.stack(
            [
                p.log_prob(imgnd_act[1:-1].detach().float()).unsqueeze(-1)
                for p, imgnd_act in zip(policies, torch.split(imagined_actions, list(actions_dim), -1))
            ],
            -1,
        ).sum(-1)
        * advantage
    )
    objective = cfg.algo.actor.objective_mix * reinforce + (1 - cfg.algo.actor.objective_mix) * dynamics
    try:
        entropy = cfg.algo.actor.ent_coef * torch.stack([p.entropy() for p in policies], -1).sum(dim=-1)
    except NotImplementedError:
        entropy = torch.zeros_like(objective)
    policy_loss = -torch.mean(discount[:-2].detach() * (objective + entropy.unsqueeze(-1)))
    runtime.backward(policy_loss)
    if cfg.algo.actor.clip_gradients and cfg.algo.actor.clip_gradients > 0:
        runtime.clip_gradients(actor, actor_optimizer, cfg.algo.actor.clip_gradients)
    actor_optimizer.step()

    # critic: Normal log-prob regression on λ-values (dreamer_v2.py:340-356)
    _qv_out = critic(imagined_trajectories.detach()[:-1]).float()

    qv = _FastUnitNormal(_qv_out, 1)
    critic_optimizer.zero_grad(set_to_none=True)
    value_loss = -torch.mean(discount[:-1, ..., 0] * qv.log_prob(lambda_values.detach()))
    runtime.backward(value_loss)
    if cfg.algo.critic.clip_gradients and cfg.algo.critic.clip_gradients > 0:
        runtime.clip_gradients(critic, critic_optimizer, cfg.algo.critic.clip_gradients)
    critic_optimizer.step()

    # hard target update every N gradient steps (dreamer_v2.py config)
    if cumulative_step % cfg.algo.critic.per_rank_target_network_update_freq == 0:
        for tp, p in zip(target_critic.parameters(), critic.parameters()):
            tp.data.copy_(p.data)

    if aggregator and not MetricAggregator.disabled:
        aggregator.update("Loss/world_model_loss", rec_loss.detach())
        aggregator.update("Loss/observation_loss", observation_loss.detach())
        aggregator.update("Loss/reward_loss", reward_loss.detach())
        aggregator.update("Loss/state_loss", state_loss.detach())
        aggregator.update("Loss/continue_loss", continue_loss.detach())
        aggregator.update("State/kl", kl.mean().detach())
        aggregator.update("Loss/policy_loss", policy_loss.detach())
        aggregator.update("Loss/value_loss", value_loss.detach())


def _capture_train_step(
    runtime, world_model, actor, critic, target_critic,
    world_optimizer, actor_optimizer, critic_optimizer,
    example_batch, cfg, is_continuous, actions_dim,
):
    """hipGraph-capture the DV2 gradient step (without the hard
    target-critic copy, which stays outside the graph on its own cadence);
    returns a replayable callable or None when capture fails."""
    from sheeprl_amd.parallel.graphs import CUDAGraphStep
    from sheeprl_amd.utils.metric import MetricAggregator

    def train_fn(batch):
        was_disabled = MetricAggregator.disabled
        MetricAggregator.disabled = True
        try:
            # cumulative_step=-1 never hits the modulo-zero target update
            train(
                runtime, world_model, actor, critic, target_critic,
                world_optimizer, actor_optimizer, critic_optimizer,
                batch, None, cfg, is_continuous, actions_dim, -1,
            )
        finally:
            MetricAggregator.disabled = was_disabled

    try:
        step = CUDAGraphStep(train_fn, example_batch, warmup=2)
        runtime.print("[dreamer_v2] gradient step captured in a hipGraph")
        return step
    except Exception as e:  # noqa: BLE001
        runtime.print(f"[dreamer_v2] hipGraph capture failed ({e}); eager training")
        return None


@register_algorithm(name="dreamer_v2")
def main(runtime: Runtime, cfg: Any) -> None:
    device = runtime.device
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
        runtime, actions_dim, is_continuous, cfg, obs_space,
        state.get("world_model"), state.get("actor"), state.get("critic"), state.get("target_critic"),
    )
    player.set_exploration(
        cfg.algo.actor.get("expl_amount", 0.0),
        cfg.algo.actor.get("expl_min", 0.0),
        cfg.algo.actor.get("expl_decay", 0.0),
    )
    world_optimizer = make_optimizer(world_model.parameters(), cfg.algo.world_model.optimizer)
    actor_optimizer = make_optimizer(actor.parameters(), cfg.algo.actor.optimizer)
    critic_optimizer = make_optimizer(critic.parameters(), cfg.algo.critic.optimizer)

    aggregator = MetricAggregator({k: "mean" for k in AGGREGATOR_KEYS})

    # buffer type: sequential | episode (dreamer_v2.py:496-517)
    buffer_size = max(int(cfg.buffer.size), 1)
    if cfg.buffer.get("type", "sequential") == "episode":
        rb: Any = EpisodeBuffer(
            buffer_size, sequence_length=cfg.algo.per_rank_sequence_length, n_envs=cfg.env.num_envs,
            obs_keys=obs_keys, prioritize_ends=cfg.buffer.get("prioritize_ends", False),
        )
    else:
        rb = EnvIndependentReplayBuffer(
            buffer_size, n_envs=cfg.env.num_envs, obs_keys=obs_keys, buffer_cls=SequentialReplayBuffer,
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
    last_checkpoint = 0
    ratio = Ratio(cfg.algo.replay_ratio, pretrain_steps=cfg.algo.per_rank_pretrain_steps)
    cumulative_steps = 0
    graphed_step = None
    want_graphs = (
        (runtime.device.type == "cuda" or os.environ.get("SHEEPRL_AMD_FORCE_GRAPHS") == "1")
        and cfg.algo.get("hip_graphs", True)
        and os.environ.get("SHEEPRL_AMD_NO_GRAPHS", "0") != "1"
    )

    import torch.nn.functional as F  # noqa: F811

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
                    real_actions = actions = np.stack(
                        [envs.single_action_space.sample() for _ in range(num_envs)]
                    )
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
                    acts = player.get_exploration_actions(torch_obs, step=policy_step)
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
            for i, ep in enumerate(infos.get("episode", [])):
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

        if isinstance(rb, EnvIndependentReplayBuffer):
            rb_ready = any(len(b) >= cfg.algo.per_rank_sequence_length for b in rb.buffer)
        else:
            rb_ready = any(ep[next(iter(ep))].shape[0] >= cfg.algo.per_rank_sequence_length for ep in rb.buffer)
        if iter_num >= learning_starts and rb_ready:
            n_steps = ratio((policy_step - prefill_steps) / world_size)
            if n_steps > 0:
                with timer("Time/train_time"):
                    for _ in range(n_steps):
                        cumulative_steps += 1
                        sample = rb.sample_tensors(
                            cfg.algo.per_rank_batch_size,
                            sequence_length=cfg.algo.per_rank_sequence_length,
                            n_samples=1,
                            device=device,
                        )
                        batch = {k: v[0].to(device) for k, v in sample.items()}
                        use_eager = graphed_step is None or (cumulative_steps % 64 == 0)  # every 64th step eager for metrics (DV3 reads the capture's static buffers instead)
                        if use_eager:
                            train(
                                runtime, world_model, actor, critic, target_critic,
                                world_optimizer, actor_optimizer, critic_optimizer,
                                batch, aggregator, cfg, is_continuous, actions_dim, cumulative_steps,
                            )
                        else:
                            graphed_step(batch)
                            # the hard target copy runs outside the graph
                            if cumulative_steps % cfg.algo.critic.per_rank_target_network_update_freq == 0:
                                with torch.no_grad():
                                    for tp, p in zip(target_critic.parameters(), critic.parameters()):
                                        tp.data.copy_(p.data)
                        if graphed_step is None and want_graphs and cumulative_steps >= 3:
                            graphed_step = _capture_train_step(
                                runtime, world_model, actor, critic, target_critic,
                                world_optimizer, actor_optimizer, critic_optimizer,
                                batch, cfg, is_continuous, actions_dim,
                            )
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
                    "actor": actor,
                    "critic": critic,
                    "target_critic": target_critic,
                    "world_optimizer": world_optimizer,
                    "actor_optimizer": actor_optimizer,
                    "critic_optimizer": critic_optimizer,
                    "ratio": ratio,
                    "policy_step": policy_step,
                    "batch_size": cfg.algo.per_rank_batch_size * world_size,
                },
                replay_buffer=rb if cfg.buffer.get("checkpoint", False) else None,
            )

    envs.close()
    if runtime.is_global_zero and cfg.algo.run_test:
        reward = test(player, runtime, make_env(cfg, cfg.seed, 0), cfg, log_dir)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()


@register_evaluation(algorithms=["dreamer_v2"])
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
    _, _, _, _, player = build_agent(
        runtime, actions_dim, is_continuous, cfg, obs_space,
        state["world_model"], state["actor"], state["critic"], state["target_critic"],
    )
    reward = test(player, runtime, env_fn, cfg)
    runtime.print(f"Test/cumulative_reward: {reward}")
    return reward
