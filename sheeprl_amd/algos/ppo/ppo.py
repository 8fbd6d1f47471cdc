"""PPO training loop (parity: sheeprl/algos/ppo/ppo.py — main :106, train :30;
optional cross-rank rollout sharing at :363-370).

Distribution model: coupled data-parallel — per-rank envs + rollout buffer,
gradient all-reduce inside ``runtime.backward`` (bucketed RCCL, overlap with
backward), identical replicas.
"""

from __future__ import annotations

import os
import time
from typing import Any, Dict

import numpy as np
import torch

from sheeprl_amd.algos.ppo.agent import build_agent
from sheeprl_amd.algos.ppo.loss import ppo_losses
from sheeprl_amd.algos.ppo.utils import AGGREGATOR_KEYS, prepare_obs, test
from sheeprl_amd.config import save_config
from sheeprl_amd.data import ReplayBuffer
from sheeprl_amd.envs import make_env, vectorize_env
from sheeprl_amd.envs import spaces
from sheeprl_amd.ops import gae as compute_gae
from sheeprl_amd.optim import make_optimizer
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.logger import get_log_dir, get_logger
from sheeprl_amd.utils.metric import MetricAggregator
from sheeprl_amd.utils.registry import register_algorithm
from sheeprl_amd.utils.timer import timer
from sheeprl_amd.utils.utils import polynomial_decay


def train(
    runtime: Runtime,
    agent: Any,
    optimizer: torch.optim.Optimizer,
    data: Dict[str, torch.Tensor],
    aggregator: MetricAggregator,
    cfg: Any,
) -> None:
    """One PPO update phase: update_epochs x minibatch SGD (parity: ppo.py:30-96)."""
    indexes = list(range(next(iter(data.values())).shape[0]))
    batch_size = cfg.algo.per_rank_batch_size
    normalize = cfg.algo.normalize_advantages
    for _ in range(cfg.algo.update_epochs):
        np.random.shuffle(indexes)
        for start in range(0, len(indexes), batch_size):
            idx = indexes[start : start + batch_size]
            batch = {k: v[idx] for k, v in data.items()}
            obs = {k: batch[k] for k in batch if k.startswith("obs_")}
            obs = {k[len("obs_") :]: v for k, v in obs.items()}
            adv = batch["advantages"]
            if normalize and adv.numel() > 1:
                adv = (adv - adv.mean()) / (adv.std() + 1e-8)
            _, logprobs, entropy, new_values = agent(obs, batch["actions"])
            pg_loss, v_loss, ent_loss = ppo_losses(
                logprobs,
                batch["logprobs"],
                adv,
                new_values.float(),  # bf16-true agent vs fp32 stored targets
                batch["values"],
                batch["returns"],
                entropy,
                cfg.algo.clip_coef,
                cfg.algo.clip_vloss,
                cfg.algo.loss_reduction,
            )
            loss = pg_loss + cfg.algo.vf_coef * v_loss + cfg.algo.ent_coef * ent_loss
            optimizer.zero_grad(set_to_none=True)
            runtime.backward(loss)
            if cfg.algo.max_grad_norm and cfg.algo.max_grad_norm > 0:
                runtime.clip_gradients(agent, optimizer, max_norm=cfg.algo.max_grad_norm)
            optimizer.step()
            if aggregator:
                aggregator.update("Loss/policy_loss", pg_loss.detach())
                aggregator.update("Loss/value_loss", v_loss.detach())
                aggregator.update("Loss/entropy_loss", ent_loss.detach())


@register_algorithm(name="ppo")
def main(runtime: Runtime, cfg: Any) -> None:
    device = runtime.device
    if cfg.buffer.share_data and runtime.world_size == 1:
        cfg.buffer.share_data = False

    # run dir + logger (rank0 creates, broadcast)
    log_dir = get_log_dir(runtime, cfg.root_dir, cfg.run_name)
    logger = get_logger(runtime, cfg, log_dir)
    runtime.logger = logger
    if runtime.is_global_zero:
        save_config(cfg, os.path.join(log_dir, "config.yaml"))

    envs = vectorize_env(cfg, cfg.seed, runtime.global_rank)
    obs_space = envs.single_observation_space
    action_space = envs.single_action_space
    if not isinstance(obs_space, spaces.Dict):
        raise RuntimeError("PPO expects dict observations (env factory dict-ifies flat spaces)")

    # resume state
    state: Dict[str, Any] = {}
    if cfg.checkpoint.resume_from:
        state = runtime.load(cfg.checkpoint.resume_from)

    agent, player = build_agent(runtime, obs_space, action_space, cfg, state.get("agent"))
    optimizer = make_optimizer(agent.parameters(), cfg.algo.optimizer)
    if "optimizer" in state:
        optimizer.load_state_dict(state["optimizer"])

    aggregator = MetricAggregator(
        {k: "mean" for k in AGGREGATOR_KEYS}, sync_on_compute=cfg.metric.get("sync_on_compute", False)
    )

    rollout_steps = cfg.algo.rollout_steps
    num_envs = cfg.env.num_envs
    world_size = runtime.world_size
    rb = ReplayBuffer(rollout_steps, num_envs, obs_keys=[f"obs_{k}" for k in obs_space.keys()])

    # step accounting (parity: howto/work_with_steps.md)
    policy_steps_per_iter = int(num_envs * world_size)
    total_steps = int(cfg.algo.total_steps)
    policy_step = int(state.get("policy_step", 0))
    last_log = int(state.get("last_log", 0))
    last_checkpoint = int(state.get("last_checkpoint", 0))
    start_iter = policy_step // (rollout_steps * policy_steps_per_iter) + 1
    num_iters = max(1, total_steps // (rollout_steps * policy_steps_per_iter)) if not cfg.dry_run else 1

    initial_ent_coef = cfg.algo.ent_coef
    initial_clip_coef = cfg.algo.clip_coef

    obs, _ = envs.reset(seed=cfg.seed + runtime.global_rank * num_envs)

    for it in range(start_iter, num_iters + 1):
        with timer("Time/env_interaction_time"):
            for _ in range(rollout_steps):
                t_obs = prepare_obs(obs, cfg, device)
                with torch.no_grad():
                    actions, logprobs, values = player.get_actions(t_obs)
                if player.actor.is_continuous:
                    env_actions = actions.cpu().numpy()
                else:
                    env_actions = actions.cpu().numpy()
                    env_actions = env_actions[..., 0] if env_actions.shape[-1] == 1 else env_actions
                next_obs, rewards, terms, truncs, infos = envs.step(env_actions)
                dones = np.logical_or(terms, truncs).astype(np.float32)

                step_data = {
                    "actions": actions.cpu().numpy().astype(np.float32)[None],
                    "logprobs": logprobs.cpu().numpy().astype(np.float32)[None],
                    "values": values.cpu().numpy().astype(np.float32)[None],
                    "rewards": rewards.astype(np.float32).reshape(1, num_envs, 1),
                    "dones": dones.reshape(1, num_envs, 1),
                }
                for k in obs_space.keys():
                    step_data[f"obs_{k}"] = np.asarray(obs[k])[None]
                rb.add(step_data)

                obs = next_obs
                policy_step += policy_steps_per_iter

                for ep in infos.get("episode", []):
                    if ep is not None:
                        aggregator.update("Rewards/rew_avg", float(ep["r"][0]))
                        aggregator.update("Game/ep_len_avg", float(ep["l"][0]))

        # bootstrap + GAE (parity: ppo.py:342-360)
        with torch.no_grad():
            t_obs = prepare_obs(obs, cfg, device)
            next_values = player.get_values(t_obs)
        local = rb.buffer
        rewards_t = torch.as_tensor(local["rewards"], device=device).float()
        values_t = torch.as_tensor(local["values"], device=device).float()
        dones_t = torch.as_tensor(local["dones"], device=device).bool()
        returns, advantages = compute_gae(
            rewards_t, values_t, dones_t, next_values, rollout_steps, cfg.algo.gamma, cfg.algo.gae_lambda
        )

        # flatten [T, n_envs, ...] -> [T*n_envs, ...]
        data: Dict[str, torch.Tensor] = {}
        for k, v in local.items():
            t = torch.as_tensor(v, device=device)
            data[k] = t.reshape(t.shape[0] * t.shape[1], *t.shape[2:])
        data["returns"] = returns.reshape(-1, 1)
        data["advantages"] = advantages.reshape(-1, 1)
        if not player.actor.is_continuous:
            data["actions"] = data["actions"].long()

        if cfg.buffer.share_data and runtime.world_size > 1:
            gathered = runtime.all_gather(data)  # [world, N, ...]
            data = {k: v.flatten(0, 1) for k, v in gathered.items()}

        with timer("Time/train_time"):
            train(runtime, agent, optimizer, data, aggregator, cfg)

        if cfg.algo.anneal_lr:
            frac = 1.0 - (it - 1) / num_iters
            for pg in optimizer.param_groups:
                pg["lr"] = cfg.algo.optimizer.lr * frac
        if cfg.algo.anneal_ent_coef:
            cfg.algo.ent_coef = polynomial_decay(it, initial=initial_ent_coef, final=0.0, max_decay_steps=num_iters)
        if cfg.algo.anneal_clip_coef:
            cfg.algo.clip_coef = polynomial_decay(it, initial=initial_clip_coef, final=0.0, max_decay_steps=num_iters)

        # logging
        if policy_step - last_log >= cfg.metric.log_every or it == num_iters or cfg.dry_run:
            metrics = aggregator.compute()
            times = timer.compute()
            if times.get("Time/train_time"):
                metrics["Time/sps_train"] = (
                    cfg.algo.update_epochs * (rollout_steps * num_envs // cfg.algo.per_rank_batch_size)
                ) / times["Time/train_time"]
            if times.get("Time/env_interaction_time"):
                metrics["Time/sps_env_interaction"] = (
                    (policy_step - last_log) / world_size * cfg.env.action_repeat
                ) / times["Time/env_interaction_time"]
            runtime.log_dict(metrics, policy_step)
            aggregator.reset()
            timer.reset()
            last_log = policy_step

        # checkpoint
        if (
            cfg.checkpoint.every > 0
            and policy_step - last_checkpoint >= cfg.checkpoint.every
            or (it == num_iters and cfg.checkpoint.save_last)
        ):
            last_checkpoint = policy_step
            ckpt_path = os.path.join(log_dir, "checkpoint", f"ckpt_{policy_step}_{runtime.global_rank}.ckpt")
            ckpt_state = {
                "agent": agent,
                "optimizer": optimizer,
                "policy_step": policy_step,
                "last_log": last_log,
                "last_checkpoint": last_checkpoint,
                "batch_size": cfg.algo.per_rank_batch_size * world_size,
            }
            runtime.call("on_checkpoint_coupled", ckpt_path=ckpt_path, state=ckpt_state)

    envs.close()
    if runtime.is_global_zero and cfg.algo.run_test:
        reward = test(player, make_env(cfg, cfg.seed, 0), cfg, log_dir, device)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()
