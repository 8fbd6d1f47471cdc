"""Plan2Explore DV1 — exploration phase.

Parity: sheeprl/algos/p2e_dv1/p2e_dv1_exploration.py — DV1 world-model
learning + disagreement ensembles over the next continuous stochastic state,
exploration actor/critic trained by dynamics backprop on the intrinsic
(λ-)returns, task actor/critic trained on the real reward.
"""

from __future__ import annotations

import os
from typing import Any, Dict

import numpy as np
import torch
import torch.distributions as td
import torch.nn.functional as F

from sheeprl_amd import ops
from sheeprl_amd.algos.dreamer_v1.dreamer_v1 import build_agent as dv1_build_agent
from sheeprl_amd.algos.dreamer_v3.agent import Actor, init_weights
from sheeprl_amd.algos.dreamer_v3.utils import prepare_obs, test
from sheeprl_amd.config import save_config
from sheeprl_amd.data import EnvIndependentReplayBuffer, SequentialReplayBuffer
from sheeprl_amd.distributions import MSEDistribution
from sheeprl_amd.envs import make_env, spaces, vectorize_env
from sheeprl_amd.models import MLP
from sheeprl_amd.optim import make_optimizer
from sheeprl_amd.algos.dreamer_v2.dreamer_v2 import _FastUnitNormal
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
}
MODELS_TO_REGISTER = {"world_model", "ensembles", "actor_task", "critic_task", "actor_exploration", "critic_exploration"}



def _unit_scale(x):
    """Scale=1 as a device tensor: td.Normal(x, 1) materializes the python
    scalar with a pageable H2D copy, which is illegal inside hipGraph capture."""
    return torch.ones((), device=x.device, dtype=x.dtype)

def _dv1_behaviour(runtime, cfg, world_model, actor, critic, actor_opt, critic_opt,
                   stochastic_states, recurrent_states, reward_fn, aggregator, tag):
    device = runtime.device
    dtype = runtime.param_dtype
    stochastic_size = cfg.algo.world_model.stochastic_size
    recurrent_state_size = cfg.algo.world_model.recurrent_model.recurrent_state_size
    horizon = cfg.algo.horizon
    flat = stochastic_states.shape[0] * stochastic_states.shape[1]

    imagined_state = stochastic_states.detach().reshape(1, -1, stochastic_size)
    recurrent_state = recurrent_states.detach().reshape(1, -1, recurrent_state_size)
    imagined_latent_states = torch.empty(horizon, flat, stochastic_size + recurrent_state_size,
                                         device=device, dtype=dtype)
    imagined_actions_list = []
    for i in range(horizon):
        actions = torch.cat(actor(torch.cat((imagined_state, recurrent_state), -1))[0], dim=-1).to(dtype)
        imagined_actions_list.append(actions)
        imagined_state, recurrent_state = world_model.rssm.imagination(imagined_state, recurrent_state, actions)
        imagined_latent_states[i] = torch.cat((imagined_state, recurrent_state), -1)
    imagined_actions = torch.cat(imagined_actions_list, dim=0)

    predicted_values = critic(imagined_latent_states).float()
    rewards = reward_fn(imagined_latent_states, imagined_actions)
    continues = torch.ones_like(rewards) * cfg.algo.gamma
    next_values = torch.cat((predicted_values[1:], predicted_values[-1:]), dim=0)
    lambda_values = ops.lambda_values(rewards, next_values, continues, cfg.algo.lmbda)
    with torch.no_grad():
        discount = torch.cumprod(torch.cat((torch.ones_like(continues[:1]), continues[:-1]), 0), 0)

    actor_opt.zero_grad(set_to_none=True)
    policy_loss = -torch.mean(discount * lambda_values)
    runtime.backward(policy_loss)
    if cfg.algo.actor.clip_gradients and cfg.algo.actor.clip_gradients > 0:
        runtime.clip_gradients(actor, actor_opt, cfg.algo.actor.clip_gradients)
    actor_opt.step()

    _qv_out = critic(imagined_latent_states.detach()).float()


    qv = _FastUnitNormal(_qv_out, 1)
    critic_opt.zero_grad(set_to_none=True)
    value_loss = -torch.mean(discount[..., 0] * qv.log_prob(lambda_values.detach()))
    runtime.backward(value_loss)
    if cfg.algo.critic.clip_gradients and cfg.algo.critic.clip_gradients > 0:
        runtime.clip_gradients(critic, critic_opt, cfg.algo.critic.clip_gradients)
    critic_opt.step()
    if aggregator and not MetricAggregator.disabled:
        aggregator.update(f"Loss/policy_loss_{tag}", policy_loss.detach())


def train(runtime, world_model, ensembles, actor_task, critic_task, actor_exploration, critic_exploration,
          world_optimizer, ensemble_optimizer, actor_task_opt, critic_task_opt, actor_expl_opt, critic_expl_opt,
          data, aggregator, cfg):
    batch_size = cfg.algo.per_rank_batch_size
    sequence_length = cfg.algo.per_rank_sequence_length
    recurrent_state_size = cfg.algo.world_model.recurrent_model.recurrent_state_size
    stochastic_size = cfg.algo.world_model.stochastic_size
    device = runtime.device
    dtype = runtime.param_dtype

    batch_obs = {k: ops.normalize_obs(data[k]).to(dtype) for k in cfg.algo.cnn_keys.encoder}
    batch_obs.update({k: data[k].to(dtype) for k in cfg.algo.mlp_keys.encoder})
    batch_actions = torch.cat((torch.zeros_like(data["actions"][:1]), data["actions"][:-1]), dim=0)

    recurrent_state = torch.zeros(1, batch_size, recurrent_state_size, device=device, dtype=dtype)
    stochastic_state = torch.zeros(1, batch_size, stochastic_size, device=device, dtype=dtype)
    recurrent_states = torch.empty(sequence_length, batch_size, recurrent_state_size, device=device, dtype=dtype)
    stochastic_states = torch.empty(sequence_length, batch_size, stochastic_size, device=device, dtype=dtype)
    post_means = torch.empty(sequence_length, batch_size, stochastic_size, device=device)
    post_stds = torch.empty(sequence_length, batch_size, stochastic_size, device=device)
    prior_means = torch.empty(sequence_length, batch_size, stochastic_size, device=device)
    prior_stds = torch.empty(sequence_length, batch_size, stochastic_size, device=device)
    embedded_obs = world_model.encoder(batch_obs)
    for i in range(sequence_length):
        recurrent_state, stochastic_state, _, post_ms, prior_ms = world_model.rssm.dynamic(
            stochastic_state, recurrent_state, batch_actions[i : i + 1], embedded_obs[i : i + 1]
        )
        recurrent_states[i] = recurrent_state
        stochastic_states[i] = stochastic_state
        post_means[i], post_stds[i] = post_ms[0], post_ms[1]
        prior_means[i], prior_stds[i] = prior_ms[0], prior_ms[1]
    latent_states = torch.cat((stochastic_states, recurrent_states), -1)

    decoded = world_model.observation_model(latent_states)
    po = {k: _FastUnitNormal(v.float(), len(v.shape[2:])) for k, v in decoded.items()}
    _rm_out = world_model.reward_model(latent_states).float()

    pr = _FastUnitNormal(_rm_out, 1)
    posteriors_dist = td.Independent(td.Normal(post_means, post_stds), 1)
    priors_dist = td.Independent(td.Normal(prior_means, prior_stds), 1)
    observation_loss = -sum(po[k].log_prob(batch_obs[k].float()).mean() for k in po)
    reward_loss = -pr.log_prob(data["rewards"]).mean()
    kl = td.kl_divergence(posteriors_dist, priors_dist).mean()
    state_loss = torch.clamp(kl, min=cfg.algo.world_model.kl_free_nats)
    rec_loss = cfg.algo.world_model.kl_regularizer * state_loss + observation_loss + reward_loss
    world_optimizer.zero_grad(set_to_none=True)
    runtime.backward(rec_loss)
    if cfg.algo.world_model.clip_gradients and cfg.algo.world_model.clip_gradients > 0:
        runtime.clip_gradients(world_model, world_optimizer, cfg.algo.world_model.clip_gradients)
    world_optimizer.step()

    # ensembles on the continuous stochastic state
    ensemble_optimizer.zero_grad(set_to_none=True)
    ens_loss = 0.0
    ens_input = torch.cat((stochastic_states.detach(), recurrent_states.detach(), data["actions"].detach().to(dtype)), -1)
    target_next = stochastic_states.detach()[1:].float()
    for ens in ensembles:
        out = ens(ens_input)[:-1].float()
        ens_loss = ens_loss - MSEDistribution(out, 1).log_prob(target_next).mean()
    runtime.backward(ens_loss)
    if cfg.algo.ensembles.clip_gradients and cfg.algo.ensembles.clip_gradients > 0:
        runtime.clip_gradients(ensembles, ensemble_optimizer, cfg.algo.ensembles.clip_gradients)
    ensemble_optimizer.step()

    def intrinsic_reward(traj, acts):
        with torch.no_grad():
            preds = torch.stack([ens(torch.cat((traj.detach(), acts.detach().view(traj.shape[0], traj.shape[1], -1)), -1)).float() for ens in ensembles])
        r = preds.var(0).mean(-1, keepdim=True) * cfg.algo.intrinsic_reward_multiplier
        if aggregator and not MetricAggregator.disabled:
            aggregator.update("Rewards/intrinsic", r.detach().mean())
        return r

    def task_reward(traj, acts):
        return world_model.reward_model(traj).float()

    _dv1_behaviour(runtime, cfg, world_model, actor_exploration, critic_exploration, actor_expl_opt,
                   critic_expl_opt, stochastic_states, recurrent_states, intrinsic_reward, aggregator,
                   "exploration")
    _dv1_behaviour(runtime, cfg, world_model, actor_task, critic_task, actor_task_opt, critic_task_opt,
                   stochastic_states, recurrent_states, task_reward, aggregator, "task")

    if aggregator and not MetricAggregator.disabled:
        aggregator.update("Loss/world_model_loss", rec_loss.detach())
        aggregator.update("Loss/ensemble_loss", ens_loss.detach())


def _capture_train_step(args, example_batch):
    """hipGraph-capture the P2E-DV1 gradient step; None on failure."""
    from sheeprl_amd.parallel.graphs import CUDAGraphStep
    from sheeprl_amd.utils.metric import MetricAggregator

    runtime, cfg = args[0], args[-1]

    def train_fn(batch):
        was_disabled = MetricAggregator.disabled
        MetricAggregator.disabled = True
        try:
            train(*args[:-1], batch, None, cfg)
        finally:
            MetricAggregator.disabled = was_disabled

    try:
        step = CUDAGraphStep(train_fn, example_batch, warmup=2)
        runtime.print("[p2e_dv1] gradient step captured in a hipGraph")
        return step
    except Exception as e:  # noqa: BLE001
        runtime.print(f"[p2e_dv1] hipGraph capture failed ({e}); eager training")
        return None


@register_algorithm(name="p2e_dv1_exploration")
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

    world_model, actor_task, critic_task, player = dv1_build_agent(
        runtime, actions_dim, is_continuous, cfg, obs_space,
        state.get("world_model"), state.get("actor_task"), state.get("critic_task"),
    )
    act = cfg.algo.get("dense_act", "elu")
    latent_state_size = cfg.algo.world_model.stochastic_size + cfg.algo.world_model.recurrent_model.recurrent_state_size
    actor_exploration = Actor(
        latent_state_size=latent_state_size,
        actions_dim=actions_dim,
        is_continuous=is_continuous,
        distribution="tanh_normal" if is_continuous else "discrete",
        init_std=cfg.algo.actor.init_std,
        min_std=cfg.algo.actor.min_std,
        dense_units=cfg.algo.actor.dense_units,
        mlp_layers=cfg.algo.actor.mlp_layers,
        unimix=0.0,
        activation=act,
        layer_norm=False,
    )
    critic_exploration = MLP(
        latent_state_size, 1, [cfg.algo.critic.dense_units] * cfg.algo.critic.mlp_layers,
        activation=act, layer_norm=False,
    )
    if state.get("actor_exploration"):
        actor_exploration.load_state_dict(state["actor_exploration"])
    if state.get("critic_exploration"):
        critic_exploration.load_state_dict(state["critic_exploration"])
    actor_exploration = runtime.setup_module(actor_exploration)
    critic_exploration = runtime.setup_module(critic_exploration)

    ens_cfg = cfg.algo.ensembles
    ensembles = torch.nn.ModuleList(
        [
            MLP(
                latent_state_size + int(np.sum(actions_dim)),
                cfg.algo.world_model.stochastic_size,
                [ens_cfg.dense_units] * ens_cfg.mlp_layers,
                activation=act,
                layer_norm=False,
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

    step_data: Dict[str, np.ndarray] = {}
    obs, _ = envs.reset(seed=cfg.seed + runtime.global_rank * num_envs)
    for k in obs_keys:
        step_data[k] = np.asarray(obs[k])[np.newaxis]
    step_data["rewards"] = np.zeros((1, num_envs, 1), dtype=np.float32)
    step_data["terminated"] = np.zeros((1, num_envs, 1), dtype=np.float32)
    step_data["truncated"] = np.zeros((1, num_envs, 1), dtype=np.float32)
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

            for ep in infos.get("episode", []):
                if ep is not None:
                    aggregator.update("Rewards/rew_avg", float(ep["r"][0]))
                    aggregator.update("Game/ep_len_avg", float(ep["l"][0]))
            for k in obs_keys:
                step_data[k] = np.asarray(next_obs[k])[np.newaxis]
            obs = next_obs
            step_data["rewards"] = np.asarray(rewards, np.float32).reshape(1, num_envs, 1)
            step_data["terminated"] = np.asarray(terminated, np.float32).reshape(1, num_envs, 1)
            step_data["truncated"] = np.asarray(truncated, np.float32).reshape(1, num_envs, 1)
            dones_idxes = np.nonzero(dones)[0].tolist()
            if dones_idxes:
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
                                runtime, world_model, ensembles, actor_task, critic_task,
                                actor_exploration, critic_exploration, world_optimizer, ensemble_optimizer,
                                actor_task_opt, critic_task_opt, actor_expl_opt, critic_expl_opt,
                                batch, aggregator, cfg,
                            )
                        if graphed_step is None and want_graphs and _n_train_calls >= 3:
                            graphed_step = _capture_train_step(
                                (runtime, world_model, ensembles, actor_task, critic_task,
                                 actor_exploration, critic_exploration, world_optimizer, ensemble_optimizer,
                                 actor_task_opt, critic_task_opt, actor_expl_opt, critic_expl_opt, cfg),
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
        reward = test(player, runtime, make_env(cfg, cfg.seed, 0), cfg, log_dir)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()


@register_evaluation(algorithms=["p2e_dv1_exploration"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    from sheeprl_amd.algos.dreamer_v1.dreamer_v1 import evaluate as dv1_eval

    mapped = dict(state)
    mapped["actor"] = state.get("actor_task")
    mapped["critic"] = state.get("critic_task")
    fn = dv1_eval.__wrapped__ if hasattr(dv1_eval, "__wrapped__") else dv1_eval
    return fn(runtime, cfg, mapped)
