"""A2C training loop (parity: sheeprl/algos/a2c/a2c.py — main :118, train :26,
losses a2c/loss.py:5/35: plain policy-gradient with GAE advantages and MSE
value loss, single pass per rollout, optional cross-rank rollout sharing
:373). Reuses the PPO agent family (reference a2c/agent.py does the same)."""

from __future__ import annotations

import os
from typing import Any, Dict

import numpy as np
import torch
import torch.nn.functional as F

from sheeprl_amd.algos.ppo.agent import build_agent
from sheeprl_amd.algos.ppo.utils import prepare_obs, test
from sheeprl_amd.config import save_config
from sheeprl_amd.data import ReplayBuffer
from sheeprl_amd.envs import make_env, vectorize_env
from sheeprl_amd.ops import gae as compute_gae
from sheeprl_amd.optim import make_optimizer
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.logger import get_log_dir, get_logger
from sheeprl_amd.utils.metric import MetricAggregator
from sheeprl_amd.utils.registry import register_algorithm, register_evaluation
from sheeprl_amd.utils.timer import timer

AGGREGATOR_KEYS = {"Rewards/rew_avg", "Game/ep_len_avg", "Loss/value_loss", "Loss/policy_loss"}
MODELS_TO_REGISTER = {"agent"}


def train(
    runtime: Runtime,
    agent: Any,
    optimizer: torch.optim.Optimizer,
    data: Dict[str, torch.Tensor],
    aggregator: MetricAggregator,
    cfg: Any,
) -> None:
    """Single-pass minibatch A2C update (parity: a2c.py:26-96)."""
    indexes = list(range(next(iter(data.values())).shape[0]))
    batch_size = cfg.algo.per_rank_batch_size or len(indexes)
    np.random.shuffle(indexes)
    for start in range(0, len(indexes), batch_size):
        idx = indexes[start : start + batch_size]
        batch = {k: v[idx] for k, v in data.items()}
        obs = {k[len("obs_") :]: v for k, v in batch.items() if k.startswith("obs_")}
        adv = batch["advantages"]
        if cfg.algo.normalize_advantages and adv.numel() > 1:
            adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        _, logprobs, entropy, values = agent(obs, batch["actions"])
        pg_loss = -(logprobs * adv).mean()
        v_loss = F.mse_loss(values, batch["returns"])
        ent_loss = -entropy.mean()
        loss = pg_loss + cfg.algo.vf_coef * v_loss + cfg.algo.ent_coef * ent_loss
        optimizer.zero_grad(set_to_none=True)
        runtime.backward(loss)
        if cfg.algo.max_grad_norm and cfg.algo.max_grad_norm > 0:
            runtime.clip_gradients(agent, optimizer, max_norm=cfg.algo.max_grad_norm)
        optimizer.step()
        if aggregator:
            aggregator.update("Loss/policy_loss", pg_loss.detach())
            aggregator.update("Loss/value_loss", v_loss.detach())


@register_algorithm(name="a2c")
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

    state: Dict[str, Any] = {}
    if cfg.checkpoint.resume_from:
        state = runtime.load(cfg.checkpoint.resume_from)

    agent, player = build_agent(runtime, obs_space, action_space, cfg, state.get("agent"))
    optimizer = make_optimizer(agent.parameters(), cfg.algo.optimizer)
    if "optimizer" in state:
        optimizer.load_state_dict(state["optimizer"])

    aggregator = MetricAggregator({k: "mean" for k in AGGREGATOR_KEYS})
    rollout_steps = cfg.algo.rollout_steps
    num_envs = cfg.env.num_envs
    world_size = runtime.world_size
    rb = ReplayBuffer(rollout_steps, num_envs, obs_keys=[f"obs_{k}" for k in obs_space.keys()])

    policy_steps_per_iter = int(num_envs * world_size)
    total_steps = int(cfg.algo.total_steps)
    policy_step = int(state.get("policy_step", 0))
    last_log = int(state.get("last_log", 0))
    last_checkpoint = int(state.get("last_checkpoint", 0))
    num_iters = max(1, total_steps // (rollout_steps * policy_steps_per_iter)) if not cfg.dry_run else 1

    obs, _ = envs.reset(seed=cfg.seed + runtime.global_rank * num_envs)

    for it in range(1, num_iters + 1):
        with timer("Time/env_interaction_time"):
            for _ in range(rollout_steps):
                t_obs = prepare_obs(obs, cfg, device)
                with torch.no_grad():
                    actions, logprobs, values = player.get_actions(t_obs)
                env_actions = actions.cpu().numpy()
                if not player.actor.is_continuous:
                    env_actions = env_actions[..., 0] if env_actions.shape[-1] == 1 else env_actions
                next_obs, rewards, terms, truncs, infos = envs.step(env_actions)
                dones = np.logical_or(terms, truncs).astype(np.float32)
                step_data = {
                    "actions": actions.cpu().numpy().astype(np.float32)[None],
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

        with torch.no_grad():
            next_values = player.get_values(prepare_obs(obs, cfg, device))
        local = rb.buffer
        returns, advantages = compute_gae(
            torch.as_tensor(local["rewards"], device=device).float(),
            torch.as_tensor(local["values"], device=device).float(),
            torch.as_tensor(local["dones"], device=device).bool(),
            next_values,
            rollout_steps,
            cfg.algo.gamma,
            cfg.algo.gae_lambda,
        )
        data: Dict[str, torch.Tensor] = {}
        for k, v in local.items():
            t = torch.as_tensor(v, device=device)
            data[k] = t.reshape(t.shape[0] * t.shape[1], *t.shape[2:])
        data["returns"] = returns.reshape(-1, 1)
        data["advantages"] = advantages.reshape(-1, 1)
        if not player.actor.is_continuous:
            data["actions"] = data["actions"].long()

        if cfg.buffer.share_data and world_size > 1:
            gathered = runtime.all_gather(data)
            data = {k: v.flatten(0, 1) for k, v in gathered.items()}

        with timer("Time/train_time"):
            train(runtime, agent, optimizer, data, aggregator, cfg)

        if cfg.algo.get("anneal_lr", False):
            frac = 1.0 - (it - 1) / num_iters
            for pg in optimizer.param_groups:
                pg["lr"] = cfg.algo.optimizer.lr * frac

        if policy_step - last_log >= cfg.metric.log_every or it == num_iters or cfg.dry_run:
            metrics = aggregator.compute()
            runtime.log_dict(metrics, policy_step)
            aggregator.reset()
            timer.reset()
            last_log = policy_step

        if (
            cfg.checkpoint.every > 0
            and policy_step - last_checkpoint >= cfg.checkpoint.every
            or (it == num_iters and cfg.checkpoint.save_last)
        ):
            last_checkpoint = policy_step
            ckpt_path = os.path.join(log_dir, "checkpoint", f"ckpt_{policy_step}_{runtime.global_rank}.ckpt")
            runtime.call(
                "on_checkpoint_coupled",
                ckpt_path=ckpt_path,
                state={
                    "agent": agent,
                    "optimizer": optimizer,
                    "policy_step": policy_step,
                    "last_log": last_log,
                    "last_checkpoint": last_checkpoint,
                    "batch_size": (cfg.algo.per_rank_batch_size or rollout_steps * num_envs) * world_size,
                },
            )

    envs.close()
    if runtime.is_global_zero and cfg.algo.run_test:
        reward = test(player, make_env(cfg, cfg.seed, 0), cfg, log_dir, device)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()


@register_evaluation(algorithms=["a2c"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    """Parity: sheeprl/algos/a2c/evaluate.py."""
    env_fn = make_env(cfg, cfg.seed, 0)
    env = env_fn()
    obs_space, action_space = env.observation_space, env.action_space
    env.close()
    _, player = build_agent(runtime, obs_space, action_space, cfg, state["agent"])
    reward = test(player, env_fn, cfg, ".", runtime.device)
    runtime.print(f"Test/cumulative_reward: {reward}")
    return reward
