"""SAC-AE training loop (parity: sheeprl/algos/sac_ae/sac_ae.py — main :120,
train :35: SAC on encoder features + autoencoder reconstruction with latent
L2, delayed actor/decoder update frequencies, separate encoder/critic EMA).

Note: the reference forces DDPStrategy(find_unused_parameters=True)
(cli.py:108-116) because the actor backward leaves encoder params unused;
GradSync handles partially-used modules natively (stragglers are zero-filled
at finalize)."""

from __future__ import annotations

import os
from typing import Any, Dict

import numpy as np
import torch
import torch.nn.functional as F

from sheeprl_amd import ops
from sheeprl_amd.algos.sac.loss import critic_loss, entropy_loss, policy_loss
from sheeprl_amd.algos.sac_ae.agent import build_agent
from sheeprl_amd.config import save_config
from sheeprl_amd.data import ReplayBuffer
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
    "Loss/value_loss",
    "Loss/policy_loss",
    "Loss/alpha_loss",
    "Loss/reconstruction_loss",
}
MODELS_TO_REGISTER = {"agent"}


def train(
    runtime: Runtime,
    agent: Any,
    encoder_optimizer, decoder_optimizer, qf_optimizer, actor_optimizer, alpha_optimizer,
    data: Dict[str, torch.Tensor],
    aggregator: MetricAggregator,
    update: int,
    cfg: Any,
) -> None:
    obs = {k: data[f"obs_{k}"] for k in data_keys(data)}
    next_obs = {k: data[f"next_obs_{k}"] for k in data_keys(data)}
    actions, rewards, not_dones = data["actions"], data["rewards"], 1.0 - data["dones"]

    # critic + encoder update
    with torch.no_grad():
        z_next_t = agent.encoder_target(next_obs)
        z_next = agent.encoder(next_obs)
        next_a, next_logp = agent.actor(z_next)
        target_qs = agent.get_target_q_values(z_next_t, next_a)
        min_t = target_qs.min(dim=-1, keepdim=True).values - agent.alpha * next_logp
        td_target = rewards + not_dones * cfg.algo.gamma * min_t
    z = agent.encoder(obs)
    qs = agent.get_q_values(z, actions)
    qf_loss = critic_loss(qs, td_target, len(agent.qfs))
    qf_optimizer.zero_grad(set_to_none=True)
    encoder_optimizer.zero_grad(set_to_none=True)
    runtime.backward(qf_loss)
    qf_optimizer.step()
    encoder_optimizer.step()
    if aggregator:
        aggregator.update("Loss/value_loss", qf_loss.detach())

    # actor + alpha (detached encoder features)
    if update % cfg.algo.actor.update_freq == 0:
        z_det = agent.encoder(obs, detach=True).detach()
        a, logp = agent.actor(z_det)
        q = agent.get_q_values(z_det, a)
        min_q = q.min(dim=-1, keepdim=True).values
        pi_loss = policy_loss(agent.alpha.detach(), logp, min_q)
        actor_optimizer.zero_grad(set_to_none=True)
        runtime.backward(pi_loss)
        actor_optimizer.step()
        a_loss = entropy_loss(agent.log_alpha, logp.detach(), agent.target_entropy)
        alpha_optimizer.zero_grad(set_to_none=True)
        a_loss.backward()
        if runtime.is_distributed and agent.log_alpha.grad is not None:
            agent.log_alpha.grad = runtime.all_reduce(agent.log_alpha.grad, op="mean")
        alpha_optimizer.step()
        if aggregator:
            aggregator.update("Loss/policy_loss", pi_loss.detach())
            aggregator.update("Loss/alpha_loss", a_loss.detach())

    # autoencoder update
    if update % cfg.algo.decoder.update_freq == 0:
        z_ae = agent.encoder(obs)
        rec = agent.decoder(z_ae)
        rec_loss = 0.0
        for k, r in rec.items():
            target = ops.normalize_obs(obs[k]) if obs[k].dtype == torch.uint8 else obs[k].float()
            rec_loss = rec_loss + F.mse_loss(r, target)
        latent_loss = 0.5 * z_ae.pow(2).sum(-1).mean()
        ae_loss = rec_loss + cfg.algo.decoder.latent_lambda * latent_loss
        encoder_optimizer.zero_grad(set_to_none=True)
        decoder_optimizer.zero_grad(set_to_none=True)
        runtime.backward(ae_loss)
        encoder_optimizer.step()
        decoder_optimizer.step()
        if aggregator:
            aggregator.update("Loss/reconstruction_loss", ae_loss.detach())

    if update % cfg.algo.critic.target_network_frequency == 0:
        agent.target_ema()


def data_keys(data: Dict[str, torch.Tensor]):
    return sorted({k[len("obs_") :] for k in data if k.startswith("obs_")})


@register_algorithm(name="sac_ae")
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
        raise RuntimeError("SAC-AE needs continuous actions")

    state: Dict[str, Any] = {}
    if cfg.checkpoint.resume_from:
        state = runtime.load(cfg.checkpoint.resume_from)

    agent, player = build_agent(runtime, cfg, obs_space, action_space, state.get("agent"))
    encoder_optimizer = make_optimizer(agent.encoder.parameters(), cfg.algo.encoder.optimizer)
    decoder_optimizer = make_optimizer(agent.decoder.parameters(), cfg.algo.decoder.optimizer)
    qf_optimizer = make_optimizer(agent.qfs.parameters(), cfg.algo.critic.optimizer)
    actor_optimizer = make_optimizer(agent.actor.parameters(), cfg.algo.actor.optimizer)
    alpha_optimizer = make_optimizer([agent.log_alpha], cfg.algo.alpha.optimizer)

    aggregator = MetricAggregator({k: "mean" for k in AGGREGATOR_KEYS})
    num_envs = cfg.env.num_envs
    world_size = runtime.world_size
    obs_keys = list(cfg.algo.cnn_keys.encoder or []) + list(cfg.algo.mlp_keys.encoder or [])
    rb = ReplayBuffer(int(cfg.buffer.size), num_envs, obs_keys=tuple(f"obs_{k}" for k in obs_keys))

    policy_steps_per_iter = int(num_envs * world_size)
    total_iters = int(cfg.algo.total_steps // policy_steps_per_iter) if not cfg.dry_run else 1
    learning_starts = cfg.algo.learning_starts // policy_steps_per_iter if not cfg.dry_run else 0
    # replay-ratio accounting starts AFTER the prefill (parity:
    # sheeprl dreamer_v3.py:661, sac.py:301 — the reference subtracts the
    # prefill policy steps before asking Ratio how many grad steps are owed,
    # otherwise the first train iteration pays a learning_starts-sized backlog)
    prefill_steps = max(learning_starts - 1, 0) * policy_steps_per_iter
    policy_step = 0
    last_log = 0
    last_checkpoint = 0
    ratio = Ratio(cfg.algo.replay_ratio, pretrain_steps=0)
    update = 0

    obs, _ = envs.reset(seed=cfg.seed + runtime.global_rank * num_envs)

    for iter_num in range(1, total_iters + 1):
        policy_step += policy_steps_per_iter
        with timer("Time/env_interaction_time"):
            if iter_num <= learning_starts:
                actions = np.stack([envs.single_action_space.sample() for _ in range(num_envs)])
            else:
                with torch.no_grad():
                    t_obs = {k: torch.as_tensor(np.asarray(obs[k]), device=device) for k in obs_keys}
                    actions = player.get_actions(t_obs).cpu().numpy()
            next_obs, rewards, terms, truncs, infos = envs.step(actions)
            dones = np.logical_or(terms, truncs).astype(np.float32)
            real_next = {k: np.asarray(next_obs[k]).copy() for k in obs_keys}
            for idx, fo in enumerate(infos.get("final_observation", [])):
                if fo is not None:
                    for k in obs_keys:
                        real_next[k][idx] = fo[k]
            step_data = {
                "actions": actions[None].astype(np.float32),
                "rewards": rewards.astype(np.float32).reshape(1, num_envs, 1),
                "dones": dones.reshape(1, num_envs, 1),
            }
            for k in obs_keys:
                step_data[f"obs_{k}"] = np.asarray(obs[k])[None]
                step_data[f"next_obs_{k}"] = real_next[k][None]
            rb.add(step_data)
            obs = next_obs
            for ep in infos.get("episode", []):
                if ep is not None:
                    aggregator.update("Rewards/rew_avg", float(ep["r"][0]))
                    aggregator.update("Game/ep_len_avg", float(ep["l"][0]))

        if iter_num >= learning_starts:
            steps = ratio((policy_step - prefill_steps) / world_size)
            if steps > 0 and len(rb) > 0:
                with timer("Time/train_time"):
                    for _ in range(steps):
                        sample = rb.sample_tensors(cfg.algo.per_rank_batch_size, n_samples=1, device=device)
                        batch = {k: v[0] for k, v in sample.items()}
                        update += 1
                        train(
                            runtime, agent, encoder_optimizer, decoder_optimizer, qf_optimizer,
                            actor_optimizer, alpha_optimizer, batch, aggregator, update, cfg,
                        )

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
                    "agent": agent,
                    "encoder_optimizer": encoder_optimizer,
                    "decoder_optimizer": decoder_optimizer,
                    "qf_optimizer": qf_optimizer,
                    "actor_optimizer": actor_optimizer,
                    "alpha_optimizer": alpha_optimizer,
                    "update": update,
                    "policy_step": policy_step,
                    "batch_size": cfg.algo.per_rank_batch_size * world_size,
                },
                replay_buffer=rb if cfg.buffer.get("checkpoint", False) else None,
            )

    envs.close()
    if runtime.is_global_zero and cfg.algo.run_test:
        reward = _test(player, make_env(cfg, cfg.seed, 0), cfg, device, obs_keys)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()


@torch.no_grad()
def _test(player, env_fn, cfg, device, obs_keys) -> float:
    env = env_fn()
    obs, _ = env.reset(seed=cfg.seed)
    done = False
    cum_reward = 0.0
    while not done:
        t_obs = {k: torch.as_tensor(np.asarray(obs[k]), device=device).unsqueeze(0) for k in obs_keys}
        action = player.get_actions(t_obs, greedy=True).cpu().numpy().reshape(-1)
        obs, reward, term, trunc, _ = env.step(action)
        cum_reward += float(reward)
        done = bool(term or trunc)
    env.close()
    return cum_reward


@register_evaluation(algorithms=["sac_ae"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    env_fn = make_env(cfg, cfg.seed, 0)
    env = env_fn()
    obs_space, action_space = env.observation_space, env.action_space
    env.close()
    _, player = build_agent(runtime, cfg, obs_space, action_space, state["agent"])
    obs_keys = list(cfg.algo.cnn_keys.encoder or []) + list(cfg.algo.mlp_keys.encoder or [])
    reward = _test(player, env_fn, cfg, runtime.device, obs_keys)
    runtime.print(f"Test/cumulative_reward: {reward}")
    return reward
