"""Decoupled SAC: rank-0 player (envs + replay buffer) + ranks 1..N-1 trainers.

Parity: sheeprl/algos/sac/sac_decoupled.py — player :89-351 (buffer held by
the player, sampled chunks scattered per training round :240-257), trainer
:356-545 (DDP over the optimization group, rank-1 broadcasts flat params
:260), shutdown sentinel -1 :318.
"""

from __future__ import annotations

import os
from typing import Any, List

import numpy as np
import torch
from sheeprl_amd.parallel import flat_to_params, params_to_flat

from sheeprl_amd.algos.sac.agent import SACAgent, SACPlayer
from sheeprl_amd.algos.sac.sac import train as sac_train
from sheeprl_amd.algos.sac.utils import AGGREGATOR_KEYS, test
from sheeprl_amd.config import save_config
from sheeprl_amd.data import ReplayBuffer
from sheeprl_amd.envs import make_env, spaces, vectorize_env
from sheeprl_amd.optim import make_optimizer
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.parallel.gradsync import GradSync
from sheeprl_amd.utils.logger import get_log_dir, get_logger
from sheeprl_amd.utils.metric import MetricAggregator
from sheeprl_amd.utils.registry import register_algorithm
from sheeprl_amd.utils.timer import timer
from sheeprl_amd.utils.utils import Ratio


def _agent_for(cfg: Any, obs_space, action_space, device) -> SACAgent:
    import sheeprl_amd.algos.sac.agent as agent_mod

    mlp_keys = list(cfg.algo.mlp_keys.encoder or [])
    obs_dim = sum(int(np.prod(obs_space[k].shape)) for k in mlp_keys)
    act_dim = int(np.prod(action_space.shape))
    actor = agent_mod.SACActor(
        obs_dim, act_dim, hidden_size=cfg.algo.actor.hidden_size,
        action_low=action_space.low, action_high=action_space.high,
    )
    critics = [
        agent_mod.SACCritic(obs_dim + act_dim, cfg.algo.critic.hidden_size, 1) for _ in range(cfg.algo.critic.n)
    ]
    return agent_mod.SACAgent(
        actor, critics, target_entropy=-act_dim, alpha=cfg.algo.alpha.alpha, tau=cfg.algo.tau, device=device
    ).to(device)


def player(runtime: Runtime, cfg: Any, world_group, pt_group) -> None:
    device = runtime.device
    log_dir = get_log_dir(runtime, cfg.root_dir, cfg.run_name, share=False)
    logger = get_logger(runtime, cfg, log_dir)
    runtime.logger = logger
    save_config(cfg, os.path.join(log_dir, "config.yaml"))

    envs = vectorize_env(cfg, cfg.seed, 0)
    obs_space = envs.single_observation_space
    action_space = envs.single_action_space
    if not isinstance(action_space, spaces.Box):
        raise RuntimeError("SAC needs continuous actions")
    mlp_keys = list(cfg.algo.mlp_keys.encoder or [])
    num_envs = cfg.env.num_envs

    agent = _agent_for(cfg, obs_space, action_space, device)
    sac_player = SACPlayer(agent.actor)
    flat = params_to_flat(agent.actor.parameters()).detach()
    runtime.broadcast(flat, src=1, group=pt_group)
    flat_to_params(flat, agent.actor.parameters())

    n_trainers = runtime.world_size - 1
    rb = ReplayBuffer(int(cfg.buffer.size), num_envs, obs_keys=("obs",))
    aggregator = MetricAggregator({k: "mean" for k in AGGREGATOR_KEYS})
    ratio = Ratio(cfg.algo.replay_ratio, pretrain_steps=0)

    total_iters = int(cfg.algo.total_steps // num_envs) if not cfg.dry_run else 1
    learning_starts = cfg.algo.learning_starts // num_envs if not cfg.dry_run else 0
    prefill_steps = max(learning_starts - 1, 0) * num_envs  # see sac.py ratio note
    policy_step = 0
    last_log = 0
    last_checkpoint = 0
    trainer_state = None
    if cfg.checkpoint.resume_from:
        _st = torch.load(cfg.checkpoint.resume_from, map_location="cpu", weights_only=False)
        last_checkpoint = int(_st.get("policy_step", 0))
        del _st

    def flat_obs(o):
        return np.concatenate(
            [np.asarray(o[k], dtype=np.float32).reshape(num_envs, -1) for k in mlp_keys], axis=-1
        )

    obs_np, _ = envs.reset(seed=cfg.seed)
    obs = flat_obs(obs_np)
    comm_dev = device if runtime.backend == "nccl" else torch.device("cpu")
    schema = None

    for iter_num in range(1, total_iters + 1):
        policy_step += num_envs
        with timer("Time/env_interaction_time"):
            if iter_num <= learning_starts:
                actions = np.stack([envs.single_action_space.sample() for _ in range(num_envs)])
            else:
                with torch.no_grad():
                    actions = sac_player.get_actions(torch.as_tensor(obs, device=device)).cpu().numpy()
            next_obs_np, rewards, terms, truncs, infos = envs.step(actions)
            dones = np.logical_or(terms, truncs).astype(np.float32)
            real_next = {k: np.asarray(next_obs_np[k]).copy() for k in mlp_keys}
            for idx, fo in enumerate(infos.get("final_observation", [])):
                if fo is not None:
                    for k in mlp_keys:
                        real_next[k][idx] = fo[k]
            rb.add(
                {
                    "obs": obs[None],
                    "next_obs": flat_obs(real_next)[None],
                    "actions": actions[None].astype(np.float32),
                    "rewards": rewards.astype(np.float32).reshape(1, num_envs, 1),
                    "dones": dones.reshape(1, num_envs, 1),
                }
            )
            obs = flat_obs(next_obs_np)
            for ep in infos.get("episode", []):
                if ep is not None:
                    aggregator.update("Rewards/rew_avg", float(ep["r"][0]))
                    aggregator.update("Game/ep_len_avg", float(ep["l"][0]))

        if iter_num >= learning_starts:
            gradient_steps = ratio(policy_step - prefill_steps)
            if gradient_steps > 0 and len(rb) > 0:
                # one scatter round per training burst: each trainer gets its
                # own stack of batches (reference :240-257)
                want_ckpt = (
                    cfg.checkpoint.every > 0
                    and policy_step - last_checkpoint >= cfg.checkpoint.every
                    or (iter_num == total_iters and cfg.checkpoint.save_last)
                )
                rows = cfg.algo.per_rank_batch_size * gradient_steps
                samples = []
                for _ in range(n_trainers):
                    s = rb.sample(rows)
                    samples.append({k: v[0] for k, v in s.items()})
                if schema is None:
                    schema = [(k, tuple(int(x) for x in samples[0][k].shape[1:])) for k in sorted(samples[0])]
                    runtime.broadcast_object_list([schema], src=0, group=world_group)
                row_w = sum(int(np.prod(sh)) for _, sh in schema)
                runtime.broadcast(
                    torch.tensor([0, rows, int(want_ckpt)], dtype=torch.int64, device=comm_dev),
                    src=0, group=world_group,
                )
                bufs = [torch.zeros(rows, row_w, dtype=torch.float32, device=comm_dev)]
                for c in samples:
                    buf = torch.empty(rows, row_w, dtype=torch.float32, device=comm_dev)
                    off = 0
                    for k, sh in schema:
                        w = int(np.prod(sh))
                        buf[:, off : off + w] = torch.as_tensor(
                            np.ascontiguousarray(c[k]).reshape(rows, w), device=comm_dev
                        )
                        off += w
                    bufs.append(buf)
                runtime.scatter_tensor(bufs[0], bufs, src=0, group=world_group)
                runtime.broadcast(flat, src=1, group=pt_group)
                flat_to_params(flat, agent.actor.parameters())
                payload: List[Any] = [None]
                runtime.broadcast_object_list(payload, src=1, group=pt_group)
                for k, v in (payload[0] or {}).items():
                    aggregator.update(k, v)
                if want_ckpt:
                    opt_payload: List[Any] = [None]
                    runtime.broadcast_object_list(opt_payload, src=1, group=pt_group)
                    trainer_state = opt_payload[0]

        if policy_step - last_log >= cfg.metric.log_every or iter_num == total_iters or cfg.dry_run:
            runtime.log_dict(aggregator.compute(), policy_step)
            aggregator.reset()
            timer.reset()
            last_log = policy_step

        if (
            cfg.checkpoint.every > 0
            and policy_step - last_checkpoint >= cfg.checkpoint.every
            or (iter_num == total_iters and cfg.checkpoint.save_last)
        ):
            last_checkpoint = policy_step
            ckpt_path = os.path.join(log_dir, "checkpoint", f"ckpt_{policy_step}_0.ckpt")
            os.makedirs(os.path.dirname(ckpt_path), exist_ok=True)
            state = {"actor": agent.actor.state_dict(), "policy_step": policy_step}
            if trainer_state is not None:
                state.update(trainer_state)  # full agent + the three optimizer states
            torch.save(state, ckpt_path, pickle_protocol=4)
            trainer_state = None

    if schema is None:
        schema = []
        runtime.broadcast_object_list([schema], src=0, group=world_group)
    runtime.broadcast(torch.tensor([1, 0, 0], dtype=torch.int64, device=comm_dev), src=0, group=world_group)
    envs.close()
    if cfg.algo.run_test:
        reward = test(sac_player, make_env(cfg, cfg.seed, 0), cfg, device)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()


def trainer(runtime: Runtime, cfg: Any, world_group, pt_group, opt_group) -> None:
    device = runtime.device
    rank = runtime.global_rank
    probe = make_env(cfg, cfg.seed, 0)()
    obs_space, action_space = probe.observation_space, probe.action_space
    probe.close()

    agent = _agent_for(cfg, obs_space, action_space, device)
    resume_state = None
    if cfg.checkpoint.resume_from and rank == 1:
        resume_state = torch.load(cfg.checkpoint.resume_from, map_location="cpu", weights_only=False)
        if "agent" in resume_state:
            agent.load_state_dict(resume_state["agent"])
    gs = GradSync(agent, bucket_cap_mb=runtime.bucket_cap_mb, process_group=opt_group)
    gs.broadcast_params(src=1)
    agent._grad_sync = gs  # noqa: SLF001
    runtime._synced_modules.append(agent)
    runtime._default_group = opt_group  # scalar all-reduces stay on the trainer group

    qf_optimizer = make_optimizer(agent.qfs.parameters(), cfg.algo.critic.optimizer)
    actor_optimizer = make_optimizer(agent.actor.parameters(), cfg.algo.actor.optimizer)
    alpha_optimizer = make_optimizer([agent.log_alpha], cfg.algo.alpha.optimizer)
    if cfg.checkpoint.resume_from:
        # rank 1 loaded the checkpoint; every trainer resumes from its states
        payload: List[Any] = [
            {k: (resume_state or {}).get(k) for k in ("qf_optimizer", "actor_optimizer", "alpha_optimizer")}
        ]
        runtime.broadcast_object_list(payload, src=1, group=opt_group)
        for name, opt in (
            ("qf_optimizer", qf_optimizer),
            ("actor_optimizer", actor_optimizer),
            ("alpha_optimizer", alpha_optimizer),
        ):
            if payload[0].get(name):
                opt.load_state_dict(payload[0][name])
    aggregator = MetricAggregator({k: "mean" for k in AGGREGATOR_KEYS})

    if rank == 1:
        runtime.broadcast(params_to_flat(agent.actor.parameters()).detach(), src=1, group=pt_group)

    update = 0
    comm_dev = device if runtime.backend == "nccl" else torch.device("cpu")
    schema = None
    while True:
        if schema is None:
            payload0: List[Any] = [None]
            runtime.broadcast_object_list(payload0, src=0, group=world_group)
            schema = payload0[0]
        ctrl = torch.zeros(3, dtype=torch.int64, device=comm_dev)
        runtime.broadcast(ctrl, src=0, group=world_group)
        if int(ctrl[0].item()) == 1:
            return
        rows = int(ctrl[1].item())
        want_ckpt = bool(int(ctrl[2].item()))
        row_w = sum(int(np.prod(sh)) for _, sh in schema)
        buf = torch.empty(rows, row_w, dtype=torch.float32, device=comm_dev)
        runtime.scatter_tensor(buf, None, src=0, group=world_group)
        if buf.device != device:
            buf = buf.to(device)
        data = {}
        off = 0
        for k, sh in schema:
            w = int(np.prod(sh))
            data[k] = buf[:, off : off + w].reshape(rows, *sh)
            off += w
        bs = cfg.algo.per_rank_batch_size
        n = rows
        for start in range(0, n, bs):
            batch = {k: v[start : start + bs] for k, v in data.items()}
            update += 1
            sac_train(
                runtime, agent, actor_optimizer, qf_optimizer, alpha_optimizer,
                batch, aggregator, update, cfg, cfg.env.num_envs,
            )
        if rank == 1:
            runtime.broadcast(params_to_flat(agent.actor.parameters()).detach(), src=1, group=pt_group)
            runtime.broadcast_object_list([aggregator.compute()], src=1, group=pt_group)
            aggregator.reset()
            if want_ckpt:
                def _cpu_sd(opt):
                    sd = opt.state_dict()
                    return {
                        "state": {
                            k: {kk: (vv.cpu() if torch.is_tensor(vv) else vv) for kk, vv in v.items()}
                            for k, v in sd["state"].items()
                        },
                        "param_groups": sd["param_groups"],
                    }

                payload = {
                    "agent": {k: v.cpu() for k, v in agent.state_dict().items()},
                    "qf_optimizer": _cpu_sd(qf_optimizer),
                    "actor_optimizer": _cpu_sd(actor_optimizer),
                    "alpha_optimizer": _cpu_sd(alpha_optimizer),
                }
                runtime.broadcast_object_list([payload], src=1, group=pt_group)


@register_algorithm(name="sac_decoupled", decoupled=True)
def main(runtime: Runtime, cfg: Any) -> None:
    if runtime.world_size < 2:
        raise RuntimeError("sac_decoupled needs at least 2 processes (1 player + >=1 trainer)")
    import torch.distributed as dist

    world_group = dist.group.WORLD
    pt_group = runtime.new_group([0, 1])
    opt_group = runtime.new_group(list(range(1, runtime.world_size)))
    if runtime.global_rank == 0:
        player(runtime, cfg, world_group, pt_group)
    else:
        trainer(runtime, cfg, world_group, pt_group, opt_group)
