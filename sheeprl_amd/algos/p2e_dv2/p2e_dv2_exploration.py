"""Plan2Explore DV2 — exploration phase.

Parity: sheeprl/algos/p2e_dv2/p2e_dv2_exploration.py — DV2 world-model
learning + disagreement ensembles (intrinsic reward = variance of the
ensemble's next-stochastic-state predictions) + dual behaviour learning
(exploration actor/critic on intrinsic reward, task actor/critic on the real
reward, both with DV2's target-critic imagination and mixed objective).
"""

from __future__ import annotations

import copy
import os
from typing import Any, Dict

import numpy as np
import torch
import torch.distributions as td

from sheeprl_amd import ops
from sheeprl_amd.algos.dreamer_v2.agent import build_agent as dv2_build_agent
from sheeprl_amd.algos.dreamer_v2.dreamer_v2 import (
    _FastUnitNormal,
    compute_lambda_values,
    dv2_reconstruction_loss,
)
from sheeprl_amd.algos.dreamer_v3.agent import init_weights
from sheeprl_amd.algos.dreamer_v3.utils import prepare_obs, test
from sheeprl_amd.config import save_config
from sheeprl_amd.data import EnvIndependentReplayBuffer, SequentialReplayBuffer
from sheeprl_amd.distributions import MSEDistribution
from sheeprl_amd.envs import make_env, spaces, vectorize_env
from sheeprl_amd.models import MLP
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
    "Loss/policy_loss_task",
    "Loss/policy_loss_exploration",
    "Loss/ensemble_loss",
    "Rewards/intrinsic",
    "State/kl",
}
MODELS_TO_REGISTER = {"world_model", "ensembles", "actor_task", "critic_task", "actor_exploration", "critic_exploration"}



def _unit_scale(x):
    """Scale=1 as a device tensor: td.Normal(x, 1) materializes the python
    scalar with a pageable H2D copy, which is illegal inside hipGraph capture."""
    return torch.ones((), device=x.device, dtype=x.dtype)

def _behaviour_update(
    runtime, cfg, world_model, actor, critic, target_critic, actor_opt, critic_opt,
    posteriors, recurrent_states, data, reward_fn, is_continuous, actions_dim, aggregator, tag,
):
    """One DV2-style behaviour-learning phase with a pluggable reward."""
    device = runtime.device
    dtype = runtime.param_dtype
    stochastic_size = cfg.algo.world_model.stochastic_size
    discrete_size = cfg.algo.world_model.discrete_size
    stoch_state_size = stochastic_size * discrete_size
    recurrent_state_size = cfg.algo.world_model.recurrent_model.recurrent_state_size
    horizon = cfg.algo.horizon
    flat = posteriors.shape[0] * posteriors.shape[1]

    imagined_prior = posteriors.detach().reshape(1, -1, stoch_state_size)
    recurrent_state = recurrent_states.detach().reshape(1, -1, recurrent_state_size)
    imagined_latent_state = torch.cat((imagined_prior, recurrent_state), -1)
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
    rewards = reward_fn(imagined_trajectories, imagined_actions)
    if cfg.algo.world_model.use_continues and world_model.continue_model:
        continues = torch.sigmoid(world_model.continue_model(imagined_trajectories).float())
        true_continue = (1 - data["terminated"]).reshape(1, -1, 1) * cfg.algo.gamma
        continues = torch.cat((true_continue, continues[1:]))
    else:
        continues = torch.ones_like(rewards.detach()) * cfg.algo.gamma

    lambda_values = compute_lambda_values(
        rewards[:-1], predicted_target_values[:-1], continues[:-1],
        bootstrap=predicted_target_values[-1:], horizon=horizon, lmbda=cfg.algo.lmbda,
    )
    with torch.no_grad():
        discount = torch.cumprod(torch.cat((torch.ones_like(continues[:1]), continues[:-1]), 0), 0)

    actor_opt.zero_grad(set_to_none=True)
    policies = actor(imagined_trajectories[:-2].detach())[1]
    dynamics = lambda_values[1:]
    advantage = (lambda_values[1:] - predicted_target_values[:-2]).detach()
    reinforce = (
        torch.stack(
            [
                p.log_prob(a[1:-1].detach().float()).unsqueeze(-1)
                for p, a in zip(policies, torch.split(imagined_actions, list(actions_dim), -1))
            ],
            -1,
        ).sum(-1)
        * advantage
    )
    objective = cfg.algo.actor.objective_mix * reinforce + (1 - cfg.algo.actor.objective_mix) * dynamics
    try:
        entropy = cfg.algo.actor.ent_coef * torch.stack([p.entropy() for p in policies], -1).sum(-1)
    except NotImplementedError:
        entropy = torch.zeros_like(objective)
    policy_loss = -torch.mean(discount[:-2].detach() * (objective + entropy.unsqueeze(-1)))
    runtime.backward(policy_loss)
    if cfg.algo.actor.clip_gradients and cfg.algo.actor.clip_gradients > 0:
        runtime.clip_gradients(actor, actor_opt, cfg.algo.actor.clip_gradients)
    actor_opt.step()

    _qv_out = critic(imagined_trajectories.detach()[:-1]).float()


    qv = _FastUnitNormal(_qv_out, 1)
    critic_opt.zero_grad(set_to_none=True)
    value_loss = -torch.mean(discount[:-1, ..., 0] * qv.log_prob(lambda_values.detach()))
    runtime.backward(value_loss)
    if cfg.algo.critic.clip_gradients and cfg.algo.critic.clip_gradients > 0:
        runtime.clip_gradients(critic, critic_opt, cfg.algo.critic.clip_gradients)
    critic_opt.step()
    if aggregator and not MetricAggregator.disabled:
        aggregator.update(f"Loss/policy_loss_{tag}", policy_loss.detach())
    return imagined_trajectories


def train(
    runtime: Runtime, world_model, ensembles, actor_task, critic_task, target_critic_task,
    actor_exploration, critic_exploration, target_critic_exploration,
    world_optimizer, ensemble_optimizer, actor_task_opt, critic_task_opt,
    actor_expl_opt, critic_expl_opt, data, aggregator, cfg, is_continuous, actions_dim, cumulative_step,
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

    # ensembles
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

    # exploration behaviour: intrinsic reward = ensemble disagreement
    def intrinsic_reward(traj, acts):
        with torch.no_grad():
            preds = torch.stack([ens(torch.cat((traj.detach(), acts.detach()), -1)).float() for ens in ensembles])
        r = preds.var(0).mean(-1, keepdim=True) * cfg.algo.intrinsic_reward_multiplier
        if aggregator and not MetricAggregator.disabled:
            aggregator.update("Rewards/intrinsic", r.detach().mean())
        return r

    _behaviour_update(
        runtime, cfg, world_model, actor_exploration, critic_exploration, target_critic_exploration,
        actor_expl_opt, critic_expl_opt, posteriors, recurrent_states, data,
        intrinsic_reward, is_continuous, actions_dim, aggregator, "exploration",
    )

    # task behaviour: real (predicted) reward
    def task_reward(traj, acts):
        return world_model.reward_model(traj).float()

    _behaviour_update(
        runtime, cfg, world_model, actor_task, critic_task, target_critic_task,
        actor_task_opt, critic_task_opt, posteriors, recurrent_states, data,
        task_reward, is_continuous, actions_dim, aggregator, "task",
    )

    if cumulative_step % cfg.algo.critic.per_rank_target_network_update_freq == 0:
        for tp, p in zip(target_critic_task.parameters(), critic_task.parameters()):
            tp.data.copy_(p.data)
        for tp, p in zip(target_critic_exploration.parameters(), critic_exploration.parameters()):
            tp.data.copy_(p.data)

    if aggregator and not MetricAggregator.disabled:
        aggregator.update("Loss/world_model_loss", rec_loss.detach())
        aggregator.update("State/kl", kl.mean().detach())
        aggregator.update("Loss/ensemble_loss", ens_loss.detach())


def _capture_train_step(args, example_batch):
    """hipGraph-capture the P2E-DV2 gradient step (hard target copies stay
    outside the graph on their own cadence); None on failure."""
    from sheeprl_amd.parallel.graphs import CUDAGraphStep
    from sheeprl_amd.utils.metric import MetricAggregator

    runtime = args[0]
    pre, post = args[:15], args[15:]  # post = (cfg, is_continuous, actions_dim)

    def train_fn(batch):
        was_disabled = MetricAggregator.disabled
        MetricAggregator.disabled = True
        try:
            train(*pre, batch, None, *post, -1)
        finally:
            MetricAggregator.disabled = was_disabled

    try:
        step = CUDAGraphStep(train_fn, example_batch, warmup=2)
        runtime.print("[p2e_dv2] gradient step captured in a hipGraph")
        return step
    except Exception as e:  # noqa: BLE001
        runtime.print(f"[p2e_dv2] hipGraph capture failed ({e}); eager training")
        return None


@register_algorithm(name="p2e_dv2_exploration")
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

    world_model, actor_task, critic_task, target_critic_task, player = dv2_build_agent(
        runtime, actions_dim, is_continuous, cfg, obs_space,
        state.get("world_model"), state.get("actor_task"), state.get("critic_task"),
        state.get("target_critic_task"),
    )
    # exploration actor/critic on top of the DV2 task agent
    from sheeprl_amd.algos.dreamer_v3.agent import Actor

    wm_cfg = cfg.algo.world_model
    latent_state_size = wm_cfg.stochastic_size * wm_cfg.discrete_size + wm_cfg.recurrent_model.recurrent_state_size
    act = cfg.algo.get("dense_act", "elu")
    actor_exploration = Actor(
        latent_state_size=latent_state_size,
        actions_dim=actions_dim,
        is_continuous=is_continuous,
        init_std=cfg.algo.actor.init_std,
        min_std=cfg.algo.actor.min_std,
        dense_units=cfg.algo.actor.dense_units,
        mlp_layers=cfg.algo.actor.mlp_layers,
        unimix=0.0,
        activation=act,
        layer_norm=bool(cfg.algo.get("layer_norm", False)),
    )
    critic_exploration = MLP(
        latent_state_size, 1, [cfg.algo.critic.dense_units] * cfg.algo.critic.mlp_layers,
        activation=act, layer_norm=bool(cfg.algo.get("layer_norm", False)),
    )
    if state.get("actor_exploration"):
        actor_exploration.load_state_dict(state["actor_exploration"])
    if state.get("critic_exploration"):
        critic_exploration.load_state_dict(state["critic_exploration"])
    actor_exploration = runtime.setup_module(actor_exploration)
    critic_exploration = runtime.setup_module(critic_exploration)
    target_critic_exploration = copy.deepcopy(critic_exploration)
    target_critic_exploration = runtime.setup_module(target_critic_exploration, sync=False)
    for p in target_critic_exploration.parameters():
        p.requires_grad_(False)

    ens_cfg = cfg.algo.ensembles
    ensembles = torch.nn.ModuleList(
        [
            MLP(
                latent_state_size + int(np.sum(actions_dim)),
                wm_cfg.stochastic_size * wm_cfg.discrete_size,
                [ens_cfg.dense_units] * ens_cfg.mlp_layers,
                activation=act,
                layer_norm=bool(cfg.algo.get("layer_norm", False)),
            )
            for _ in range(ens_cfg.n)
        ]
    )
    for i, ens in enumerate(ensembles):
        torch.manual_seed(cfg.seed + runtime.global_rank * 1000 + i)
        ens.apply(init_weights)
    if state.get("ensembles"):
        ensembles.load_state_dict(state["ensembles"])
    ensembles = runtime.setup_module(ensembles)

    player.actor = actor_exploration
    player.actor_type = "exploration"

    world_optimizer = make_optimizer(world_model.parameters(), cfg.algo.world_model.optimizer)
    ensemble_optimizer = make_optimizer(ensembles.parameters(), cfg.algo.ensembles.optimizer)
    actor_task_opt = make_optimizer(actor_task.parameters(), cfg.algo.actor.optimizer)
    critic_task_opt = make_optimizer(critic_task.parameters(), cfg.algo.critic.optimizer)
    actor_expl_opt = make_optimizer(actor_exploration.parameters(), cfg.algo.actor.optimizer)
    critic_expl_opt = make_optimizer(critic_exploration.parameters(), cfg.algo.critic.optimizer)

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
    cumulative_steps = 0

    import torch.nn.functional as F

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
                        cumulative_steps += 1
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
                            if cumulative_steps % cfg.algo.critic.per_rank_target_network_update_freq == 0:
                                with torch.no_grad():
                                    for tp, p_ in zip(target_critic_task.parameters(), critic_task.parameters()):
                                        tp.data.copy_(p_.data)
                                    for tp, p_ in zip(target_critic_exploration.parameters(),
                                                      critic_exploration.parameters()):
                                        tp.data.copy_(p_.data)
                        else:
                            train(
                                runtime, world_model, ensembles, actor_task, critic_task, target_critic_task,
                                actor_exploration, critic_exploration, target_critic_exploration,
                                world_optimizer, ensemble_optimizer, actor_task_opt, critic_task_opt,
                                actor_expl_opt, critic_expl_opt, batch, aggregator, cfg,
                                is_continuous, actions_dim, cumulative_steps,
                            )
                        if graphed_step is None and want_graphs and _n_train_calls >= 3:
                            graphed_step = _capture_train_step(
                                (runtime, world_model, ensembles, actor_task, critic_task, target_critic_task,
                                 actor_exploration, critic_exploration, target_critic_exploration,
                                 world_optimizer, ensemble_optimizer, actor_task_opt, critic_task_opt,
                                 actor_expl_opt, critic_expl_opt, cfg, is_continuous, actions_dim),
                                batch,
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
                    "ensembles": ensembles,
                    "actor_task": actor_task,
                    "critic_task": critic_task,
                    "target_critic_task": target_critic_task,
                    "actor_exploration": actor_exploration,
                    "critic_exploration": critic_exploration,
                    "ratio": ratio,
                    "policy_step": policy_step,
                    "batch_size": cfg.algo.per_rank_batch_size * world_size,
                },
                replay_buffer=rb if cfg.buffer.get("checkpoint", False) else None,
            )

    envs.close()
    if runtime.is_global_zero and cfg.algo.run_test:
        player.actor = actor_task
        player.actor_type = "task"
        reward = test(player, runtime, make_env(cfg, cfg.seed, 0), cfg, log_dir)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()


@register_evaluation(algorithms=["p2e_dv2_exploration"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    from sheeprl_amd.algos.dreamer_v2.dreamer_v2 import evaluate as dv2_eval

    mapped = dict(state)
    mapped["actor"] = state.get("actor_task")
    mapped["critic"] = state.get("critic_task")
    mapped["target_critic"] = state.get("target_critic_task") or state.get("critic_task")
    return dv2_eval.__wrapped__(runtime, cfg, mapped) if hasattr(dv2_eval, "__wrapped__") else dv2_eval(runtime, cfg, mapped)
