"""Recurrent PPO training loop (parity: sheeprl/algos/ppo_recurrent/
ppo_recurrent.py — main :120, train :31: BPTT over rollout sequences with
stored initial LSTM states; sequence minibatches over the env dimension)."""

from __future__ import annotations

import os
from typing import Any, Dict

import numpy as np
import torch

from sheeprl_amd.algos.ppo.loss import ppo_losses
from sheeprl_amd.algos.ppo.utils import AGGREGATOR_KEYS, prepare_obs
from sheeprl_amd.algos.ppo_recurrent.agent import build_agent
from sheeprl_amd.config import save_config
from sheeprl_amd.envs import make_env, vectorize_env
from sheeprl_amd.ops import gae as compute_gae
from sheeprl_amd.optim import make_optimizer
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.logger import get_log_dir, get_logger
from sheeprl_amd.utils.metric import MetricAggregator
from sheeprl_amd.utils.registry import register_algorithm, register_evaluation
from sheeprl_amd.utils.timer import timer
from sheeprl_amd.utils.utils import polynomial_decay
MODELS_TO_REGISTER = {"agent"}


def _onehot_actions(actions: torch.Tensor, actions_dim, continuous: bool) -> torch.Tensor:
    if continuous:
        return actions.float()
    parts = []
    for i, n in enumerate(actions_dim):
        parts.append(torch.nn.functional.one_hot(actions[..., i].long(), n).float())
    return torch.cat(parts, dim=-1)


@register_algorithm(name="ppo_recurrent")
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
    obs_keys = list(obs_space.keys())
    act_dim_sum = int(np.sum(agent.actions_dim)) if not agent.is_continuous else agent.actions_dim[0]

    policy_steps_per_iter = int(num_envs * world_size)
    total_steps = int(cfg.algo.total_steps)
    policy_step = int(state.get("policy_step", 0))
    last_log = 0
    last_checkpoint = 0
    num_iters = max(1, total_steps // (rollout_steps * policy_steps_per_iter)) if not cfg.dry_run else 1

    obs, _ = envs.reset(seed=cfg.seed + runtime.global_rank * num_envs)
    prev_actions = torch.zeros(1, num_envs, act_dim_sum, device=device)
    states = agent.initial_states(num_envs, device)
    is_first_np = np.ones((num_envs,), dtype=np.float32)

    initial_ent_coef = float(cfg.algo.ent_coef)
    initial_clip_coef = float(cfg.algo.clip_coef)
    for it in range(1, num_iters + 1):
        rollout: Dict[str, list] = {k: [] for k in
                                    ["actions", "prev_actions", "logprobs", "values", "rewards", "dones", "is_first"]}
        for k in obs_keys:
            rollout[f"obs_{k}"] = []
        init_states = (states[0].detach().clone(), states[1].detach().clone())

        with timer("Time/env_interaction_time"):
            for _ in range(rollout_steps):
                t_obs = {k: v.unsqueeze(0) for k, v in prepare_obs(obs, cfg, device).items()}
                # reset recurrent states for envs that restarted
                # (reference ppo_recurrent.yaml: reset_recurrent_state_on_done)
                mask = torch.as_tensor(1.0 - is_first_np, device=device).view(1, -1, 1).float()
                if cfg.algo.get("reset_recurrent_state_on_done", True):
                    states = (states[0] * mask, states[1] * mask)
                prev_actions = prev_actions * mask
                with torch.no_grad():
                    actions, logprobs, values, states = player.get_actions(t_obs, prev_actions, states)
                env_actions = actions.squeeze(0).cpu().numpy()
                if not agent.is_continuous:
                    env_actions = env_actions[..., 0] if env_actions.shape[-1] == 1 else env_actions
                next_obs, rewards, terms, truncs, infos = envs.step(env_actions)
                dones = np.logical_or(terms, truncs).astype(np.float32)

                rollout["actions"].append(actions.squeeze(0).float().cpu().numpy())
                rollout["prev_actions"].append(prev_actions.squeeze(0).cpu().numpy())
                rollout["logprobs"].append(logprobs.squeeze(0).cpu().numpy())
                rollout["values"].append(values.squeeze(0).cpu().numpy())
                rollout["rewards"].append(rewards.astype(np.float32).reshape(num_envs, 1))
                rollout["dones"].append(dones.reshape(num_envs, 1))
                rollout["is_first"].append(is_first_np.reshape(num_envs, 1).copy())
                for k in obs_keys:
                    rollout[f"obs_{k}"].append(np.asarray(obs[k]))

                prev_actions = _onehot_actions(actions, agent.actions_dim, agent.is_continuous).to(device)
                is_first_np = dones.copy()
                obs = next_obs
                policy_step += policy_steps_per_iter
                for ep in infos.get("episode", []):
                    if ep is not None:
                        aggregator.update("Rewards/rew_avg", float(ep["r"][0]))
                        aggregator.update("Game/ep_len_avg", float(ep["l"][0]))

        data = {k: torch.as_tensor(np.stack(v), device=device) for k, v in rollout.items()}
        with torch.no_grad():
            t_obs = {k: v.unsqueeze(0) for k, v in prepare_obs(obs, cfg, device).items()}
            _, _, next_values, _ = player.get_actions(t_obs, prev_actions, states)
            next_values = next_values.squeeze(0)
        returns, advantages = compute_gae(
            data["rewards"].float(),
            data["values"].float(),
            data["dones"].bool(),
            next_values,
            rollout_steps,
            cfg.algo.gamma,
            cfg.algo.gae_lambda,
        )
        data["returns"] = returns
        data["advantages"] = advantages

        # sequence minibatches over the env axis (BPTT through the rollout)
        with timer("Time/train_time"):
            env_idxs = np.arange(num_envs)
            bs = max(1, min(cfg.algo.per_rank_batch_size, num_envs))
            for _ in range(cfg.algo.update_epochs):
                np.random.shuffle(env_idxs)
                for start in range(0, num_envs, bs):
                    sel = env_idxs[start : start + bs]
                    batch_obs = {k[len("obs_") :]: data[k][:, sel] for k in data if k.startswith("obs_")}
                    actions = data["actions"][:, sel]
                    if not agent.is_continuous:
                        actions = actions.long()
                    h0 = init_states[0][:, sel].contiguous()
                    c0 = init_states[1][:, sel].contiguous()
                    logp, ent, values_new = agent.forward_sequence(
                        batch_obs, data["prev_actions"][:, sel], data["is_first"][:, sel], (h0, c0), actions
                    )
                    adv = data["advantages"][:, sel]
                    if cfg.algo.normalize_advantages and adv.numel() > 1:
                        adv = (adv - adv.mean()) / (adv.std() + 1e-8)
                    pg, vl, el = ppo_losses(
                        logp, data["logprobs"][:, sel], adv, values_new, data["values"][:, sel],
                        data["returns"][:, sel], ent, cfg.algo.clip_coef, cfg.algo.clip_vloss,
                        cfg.algo.loss_reduction,
                    )
                    loss = pg + cfg.algo.vf_coef * vl + cfg.algo.ent_coef * el
                    optimizer.zero_grad(set_to_none=True)
                    runtime.backward(loss)
                    if cfg.algo.max_grad_norm and cfg.algo.max_grad_norm > 0:
                        runtime.clip_gradients(agent, optimizer, max_norm=cfg.algo.max_grad_norm)
                    optimizer.step()
                    aggregator.update("Loss/policy_loss", pg.detach())
                    aggregator.update("Loss/value_loss", vl.detach())
                    aggregator.update("Loss/entropy_loss", el.detach())


        if cfg.algo.get("anneal_lr", False):
            frac = 1.0 - (it - 1) / num_iters
            for pg in optimizer.param_groups:
                pg["lr"] = cfg.algo.optimizer.lr * frac
        if cfg.algo.get("anneal_ent_coef", False):
            cfg.algo.ent_coef = polynomial_decay(it, initial=initial_ent_coef, final=0.0, max_decay_steps=num_iters)
        if cfg.algo.get("anneal_clip_coef", False):
            cfg.algo.clip_coef = polynomial_decay(it, initial=initial_clip_coef, final=0.0, max_decay_steps=num_iters)
        if policy_step - last_log >= cfg.metric.log_every or it == num_iters or cfg.dry_run:
            runtime.log_dict(aggregator.compute(), policy_step)
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
                state={"agent": agent, "optimizer": optimizer, "policy_step": policy_step,
                       "batch_size": cfg.algo.per_rank_batch_size * world_size},
            )

    envs.close()
    if runtime.is_global_zero and cfg.algo.run_test:
        reward = _test(player, make_env(cfg, cfg.seed, 0), cfg, device, agent)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()


@torch.no_grad()
def _test(player, env_fn, cfg, device, agent) -> float:
    env = env_fn()
    obs, _ = env.reset(seed=cfg.seed)
    done = False
    cum_reward = 0.0
    act_dim_sum = int(np.sum(agent.actions_dim)) if not agent.is_continuous else agent.actions_dim[0]
    prev_actions = torch.zeros(1, 1, act_dim_sum, device=device)
    states = agent.initial_states(1, device)
    while not done:
        batched = {k: np.expand_dims(np.asarray(v), 0) for k, v in obs.items()}
        t_obs = {k: v.unsqueeze(0) for k, v in prepare_obs(batched, cfg, device).items()}
        actions, _, _, states = player.get_actions(t_obs, prev_actions, states, greedy=True)
        prev_actions = _onehot_actions(actions, agent.actions_dim, agent.is_continuous).to(device)
        a = actions.cpu().numpy().reshape(-1)
        if not agent.is_continuous:
            a = a[0] if a.shape[0] == 1 else a
        obs, reward, term, trunc, _ = env.step(a)
        cum_reward += float(reward)
        done = bool(term or trunc)
    env.close()
    return cum_reward


@register_evaluation(algorithms=["ppo_recurrent"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    env_fn = make_env(cfg, cfg.seed, 0)
    env = env_fn()
    obs_space, action_space = env.observation_space, env.action_space
    env.close()
    agent, player = build_agent(runtime, obs_space, action_space, cfg, state["agent"])
    reward = _test(player, env_fn, cfg, runtime.device, agent)
    runtime.print(f"Test/cumulative_reward: {reward}")
    return reward
