"""SAC training loop (parity: sheeprl/algos/sac/sac.py — main :82, train :32;
cross-rank sample gather :306-337, scalar alpha-grad all-reduce :72).

Off-policy coupled DP: per-rank envs + replay buffer; each rank samples
locally, the samples are all-gathered and partitioned so every rank trains on
the global sample (the reference's DistributedSampler trick); gradient
all-reduce through GradSync; the entropy-coefficient gradient is a scalar
all-reduce.
"""

from __future__ import annotations

import os
from typing import Any, Dict

import numpy as np
import torch

from sheeprl_amd.algos.sac.agent import build_agent
from sheeprl_amd.algos.sac.loss import critic_loss, entropy_loss, policy_loss
from sheeprl_amd.algos.sac.utils import AGGREGATOR_KEYS, test
from sheeprl_amd.config import save_config
from sheeprl_amd.data import ReplayBuffer
from sheeprl_amd.envs import make_env, spaces, vectorize_env
from sheeprl_amd.optim import make_optimizer
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.logger import get_log_dir, get_logger
from sheeprl_amd.utils.metric import MetricAggregator
from sheeprl_amd.utils.registry import register_algorithm
from sheeprl_amd.utils.timer import timer
from sheeprl_amd.utils.utils import Ratio


def train(
    runtime: Runtime,
    agent: Any,
    actor_optimizer: torch.optim.Optimizer,
    qf_optimizer: torch.optim.Optimizer,
    alpha_optimizer: torch.optim.Optimizer,
    data: Dict[str, torch.Tensor],
    aggregator: MetricAggregator,
    update: int,
    cfg: Any,
    policy_steps_per_iter: int,
) -> None:
    num_critics = len(agent.qfs)
    obs = data["obs"]
    next_obs = data["next_obs"]
    actions = data["actions"]
    rewards = data["rewards"]
    not_dones = 1.0 - data["dones"]

    # critic update
    with torch.no_grad():
        next_a, next_logp = agent.actor(next_obs)
        target_qs = agent.get_target_q_values(next_obs, next_a)
        min_target = target_qs.min(dim=-1, keepdim=True).values - agent.alpha * next_logp
        next_qf_value = rewards + not_dones * cfg.algo.gamma * min_target
    qs = agent.get_q_values(obs, actions)
    qf_loss = critic_loss(qs, next_qf_value.to(qs.dtype), num_critics)
    qf_optimizer.zero_grad(set_to_none=True)
    runtime.backward(qf_loss)
    qf_optimizer.step()
    if aggregator:
        aggregator.update("Loss/value_loss", qf_loss.detach())

    # actor update (possibly delayed)
    if update % cfg.algo.actor.update_freq == 0:
        for _ in range(cfg.algo.actor.network_frequency if hasattr(cfg.algo.actor, "network_frequency") else 1):
            a, logp = agent.actor(obs)
            q = agent.get_q_values(obs, a)
            min_q = q.min(dim=-1, keepdim=True).values
            pi_loss = policy_loss(agent.alpha.detach(), logp, min_q)
            actor_optimizer.zero_grad(set_to_none=True)
            runtime.backward(pi_loss)
            actor_optimizer.step()

            # alpha update with scalar gradient all-reduce (reference sac.py:72)
            a_loss = entropy_loss(agent.log_alpha, logp.detach(), agent.target_entropy)
            alpha_optimizer.zero_grad(set_to_none=True)
            a_loss.backward()
            if runtime.is_distributed and agent.log_alpha.grad is not None:
                agent.log_alpha.grad = runtime.all_reduce(agent.log_alpha.grad, op="mean")
            alpha_optimizer.step()
            if aggregator:
                aggregator.update("Loss/policy_loss", pi_loss.detach())
                aggregator.update("Loss/alpha_loss", a_loss.detach())

    # target EMA (possibly delayed)
    if update % cfg.algo.critic.target_network_frequency == 0:
        agent.qfs_target_ema()


@register_algorithm(name="sac")
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
    if not isinstance(action_space, spaces.Box):
        raise RuntimeError(f"SAC needs continuous actions, got {action_space}")

    state: Dict[str, Any] = {}
    if cfg.checkpoint.resume_from:
        state = runtime.load(cfg.checkpoint.resume_from)

    agent, player = build_agent(runtime, cfg, obs_space, action_space, state.get("agent"))
    qf_optimizer = make_optimizer(agent.qfs.parameters(), cfg.algo.critic.optimizer)
    actor_optimizer = make_optimizer(agent.actor.parameters(), cfg.algo.actor.optimizer)
    alpha_optimizer = make_optimizer([agent.log_alpha], cfg.algo.alpha.optimizer)
    for name, opt in (("qf_optimizer", qf_optimizer), ("actor_optimizer", actor_optimizer),
                      ("alpha_optimizer", alpha_optimizer)):
        if name in state:
            opt.load_state_dict(state[name])

    aggregator = MetricAggregator(
        {k: "mean" for k in AGGREGATOR_KEYS}, sync_on_compute=cfg.metric.get("sync_on_compute", False)
    )

    num_envs = cfg.env.num_envs
    world_size = runtime.world_size
    mlp_keys = list(cfg.algo.mlp_keys.encoder or [])
    rb = ReplayBuffer(
        int(cfg.buffer.size),
        num_envs,
        obs_keys=("obs",),
        memmap=cfg.buffer.memmap,
        memmap_dir=os.path.join(log_dir, "memmap_buffer", f"rank_{runtime.global_rank}"),
    )
    if "rb" in state and state["rb"] is not None:
        rbs = state["rb"]
        rb.load_state_dict(rbs[runtime.global_rank % len(rbs)] if isinstance(rbs, list) else rbs)

    policy_steps_per_iter = int(num_envs * world_size)
    total_iters = int(cfg.algo.total_steps // policy_steps_per_iter) if not cfg.dry_run else 1
    learning_starts = cfg.algo.learning_starts // policy_steps_per_iter if not cfg.dry_run else 0
    # replay-ratio accounting starts AFTER the prefill (parity:
    # sheeprl dreamer_v3.py:661, sac.py:301 — the reference subtracts the
    # prefill policy steps before asking Ratio how many grad steps are owed,
    # otherwise the first train iteration pays a learning_starts-sized backlog)
    prefill_steps = max(learning_starts - 1, 0) * policy_steps_per_iter
    policy_step = int(state.get("policy_step", 0))
    last_log = int(state.get("last_log", 0))
    last_checkpoint = int(state.get("last_checkpoint", 0))
    start_iter = int(state.get("iter_num", 1))
    ratio = Ratio(cfg.algo.replay_ratio, pretrain_steps=0)
    if "ratio" in state:
        ratio.load_state_dict(state["ratio"])
    update_counter = int(state.get("update", 0))

    obs_np, _ = envs.reset(seed=cfg.seed + runtime.global_rank * num_envs)

    def flat_obs(o):
        arrs = [np.asarray(o[k], dtype=np.float32).reshape(num_envs, -1) for k in mlp_keys]
        return np.concatenate(arrs, axis=-1)

    obs = flat_obs(obs_np)

    for iter_num in range(start_iter, total_iters + 1):
        policy_step += policy_steps_per_iter
        with timer("Time/env_interaction_time"):
            if iter_num <= learning_starts and cfg.checkpoint.resume_from is None:
                actions = np.stack([envs.single_action_space.sample() for _ in range(num_envs)])
            else:
                with torch.no_grad():
                    t_obs = torch.as_tensor(obs, device=device, dtype=torch.float32)
                    actions = player.get_actions(t_obs).cpu().numpy()
            next_obs_np, rewards, terms, truncs, infos = envs.step(actions)
            dones = np.logical_or(terms, truncs).astype(np.float32)
            # use the true final obs for the stored transition
            real_next = {k: np.asarray(next_obs_np[k]).copy() for k in mlp_keys}
            for idx, final_obs in enumerate(infos.get("final_observation", [])):
                if final_obs is not None:
                    for k in mlp_keys:
                        real_next[k][idx] = final_obs[k]
            next_obs = flat_obs(real_next)
            step_data = {
                "obs": obs[None],
                "next_obs": next_obs[None],
                "actions": actions[None].astype(np.float32),
                "rewards": rewards.astype(np.float32).reshape(1, num_envs, 1),
                "dones": dones.reshape(1, num_envs, 1),
            }
            rb.add(step_data)
            obs = flat_obs(next_obs_np)

            for ep in infos.get("episode", []):
                if ep is not None:
                    aggregator.update("Rewards/rew_avg", float(ep["r"][0]))
                    aggregator.update("Game/ep_len_avg", float(ep["l"][0]))

        if iter_num >= learning_starts:
            per_rank_gradient_steps = ratio((policy_step - prefill_steps) / world_size)
            if per_rank_gradient_steps > 0 and len(rb) >= 1:
                with timer("Time/train_time"):
                    # sample all batches at once, share across ranks (sac.py:306-337)
                    sample = rb.sample_tensors(
                        cfg.algo.per_rank_batch_size * per_rank_gradient_steps,
                        sample_next_obs=False,
                        n_samples=1,
                        device=device,
                    )
                    local = {k: v[0] for k, v in sample.items()}
                    if world_size > 1:
                        gathered = runtime.all_gather(local)  # [world, N, ...]
                        full = {k: v.flatten(0, 1) for k, v in gathered.items()}
                        # partition: this rank trains on its shard of the global sample
                        n = full["obs"].shape[0]
                        idx = torch.arange(runtime.global_rank, n, world_size, device=device)
                        local = {k: v[idx] for k, v in full.items()}
                    bs = cfg.algo.per_rank_batch_size
                    for g in range(per_rank_gradient_steps):
                        batch = {k: v[g * bs : (g + 1) * bs] for k, v in local.items()}
                        if batch["obs"].shape[0] == 0:
                            break
                        update_counter += 1
                        train(
                            runtime, agent, actor_optimizer, qf_optimizer, alpha_optimizer,
                            batch, aggregator, update_counter, cfg, policy_steps_per_iter,
                        )

        if policy_step - last_log >= cfg.metric.log_every or iter_num == total_iters or cfg.dry_run:
            metrics = aggregator.compute()
            times = timer.compute()
            if times.get("Time/train_time"):
                metrics["Time/sps_train"] = update_counter * world_size / times["Time/train_time"]
            if times.get("Time/env_interaction_time"):
                metrics["Time/sps_env_interaction"] = (
                    ((policy_step - last_log) / world_size) * cfg.env.action_repeat
                ) / times["Time/env_interaction_time"]
            runtime.log_dict(metrics, policy_step)
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
            ckpt_state = {
                "agent": agent,
                "qf_optimizer": qf_optimizer,
                "actor_optimizer": actor_optimizer,
                "alpha_optimizer": alpha_optimizer,
                "ratio": ratio,
                "update": update_counter,
                "iter_num": iter_num + 1,
                "policy_step": policy_step,
                "last_log": last_log,
                "last_checkpoint": last_checkpoint,
                "batch_size": cfg.algo.per_rank_batch_size * world_size,
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
        reward = test(player, make_env(cfg, cfg.seed, 0), cfg, device)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()
