"""Decoupled PPO: rank-0 player + ranks 1..N-1 trainers.

Parity: sheeprl/algos/ppo/ppo_decoupled.py — player :33, trainer :368,
main :624; process groups :645-664 (world / player<->trainer [0,1] /
optimization group 1..N-1), flat-parameter weight sync :119-127/:302-305/
:421-424/:551-554, rollout chunk scatter :294-299, shutdown sentinel -1
:344/:463, metrics broadcast :310/:578.

MI355X mapping: the player rank pins env interaction +
inference on its GPU; trainers DDP-train over an RCCL communicator spanning
ranks 1..N-1; the flat-parameter broadcast player<->rank-1 rides one xGMI
link.  Rollout chunks travel as ONE device tensor per trainer
(``Runtime.scatter_tensor`` — RCCL point-to-point over xGMI; the reference
pickles python objects), with a 2-int control broadcast carrying the
data/shutdown flag and the true row count.  Chunks are UNEVEN (reference
semantics): trainers that exhaust their rows early run Join-equivalent
zero-gradient sync rounds (``GradSync.sync_zero``) so bucketed all-reduce
call counts stay matched while replicas remain bit-identical.  Objects are
only used for the low-rate control plane (schema/metrics/checkpoints).
"""

from __future__ import annotations

import os
from typing import Any, Dict, List

import numpy as np
import torch
from sheeprl_amd.parallel import flat_to_params, params_to_flat

from sheeprl_amd.algos.ppo.agent import PPOAgent, PPOPlayer
from sheeprl_amd.algos.ppo.loss import ppo_losses
from sheeprl_amd.algos.ppo.utils import AGGREGATOR_KEYS, prepare_obs, test
from sheeprl_amd.config import save_config
from sheeprl_amd.data import ReplayBuffer
from sheeprl_amd.envs import make_env, spaces, vectorize_env
from sheeprl_amd.ops import gae as compute_gae
from sheeprl_amd.optim import make_optimizer
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.parallel.gradsync import GradSync
from sheeprl_amd.utils.logger import get_log_dir, get_logger
from sheeprl_amd.utils.metric import MetricAggregator
from sheeprl_amd.utils.registry import register_algorithm
from sheeprl_amd.utils.timer import timer


def player(runtime: Runtime, cfg: Any, world_group, pt_group) -> None:
    device = runtime.device
    log_dir = get_log_dir(runtime, cfg.root_dir, cfg.run_name, share=False)
    logger = get_logger(runtime, cfg, log_dir)
    runtime.logger = logger
    save_config(cfg, os.path.join(log_dir, "config.yaml"))

    envs = vectorize_env(cfg, cfg.seed, 0)
    obs_space = envs.single_observation_space
    action_space = envs.single_action_space

    agent = PPOAgent(obs_space, action_space, cfg.algo).to(device)
    ppo_player = PPOPlayer(agent.feature_extractor, agent.actor, agent.critic)

    # initial weights come from the lead trainer (rank 1)
    flat = params_to_flat(agent.parameters()).detach()
    runtime.broadcast(flat, src=1, group=pt_group)
    flat_to_params(flat, agent.parameters())

    n_trainers = runtime.world_size - 1
    rollout_steps = cfg.algo.rollout_steps
    num_envs = cfg.env.num_envs
    rb = ReplayBuffer(rollout_steps, num_envs, obs_keys=[f"obs_{k}" for k in obs_space.keys()])
    aggregator = MetricAggregator({k: "mean" for k in AGGREGATOR_KEYS})

    total_steps = int(cfg.algo.total_steps)
    policy_step = 0
    if cfg.checkpoint.resume_from:
        _st = torch.load(cfg.checkpoint.resume_from, map_location="cpu", weights_only=False)
        policy_step = int(_st.get("policy_step", 0))
        del _st
    last_log = policy_step
    last_checkpoint = policy_step
    num_iters = max(1, total_steps // (rollout_steps * num_envs)) if not cfg.dry_run else 1

    # tensor data plane setup: collectives ride RCCL on GPU, gloo on CPU
    comm_dev = device if runtime.backend == "nccl" else torch.device("cpu")
    schema = None

    obs, _ = envs.reset(seed=cfg.seed)
    for it in range(1, num_iters + 1):
        with timer("Time/env_interaction_time"):
            for _ in range(rollout_steps):
                t_obs = prepare_obs(obs, cfg, device)
                with torch.no_grad():
                    actions, logprobs, values = ppo_player.get_actions(t_obs)
                env_actions = actions.cpu().numpy()
                if not ppo_player.actor.is_continuous:
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
                policy_step += num_envs
                for ep in infos.get("episode", []):
                    if ep is not None:
                        aggregator.update("Rewards/rew_avg", float(ep["r"][0]))
                        aggregator.update("Game/ep_len_avg", float(ep["l"][0]))

        with torch.no_grad():
            next_values = ppo_player.get_values(prepare_obs(obs, cfg, device))
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
        data: Dict[str, np.ndarray] = {}
        for k, v in local.items():
            data[k] = np.asarray(v).reshape(v.shape[0] * v.shape[1], *v.shape[2:])
        data["returns"] = returns.reshape(-1, 1).cpu().numpy()
        data["advantages"] = advantages.reshape(-1, 1).cpu().numpy()

        # tensor data plane: uneven contiguous chunks packed into one flat
        # fp32 tensor per trainer (equal buffer sizes for the scatter; each
        # trainer slices its true row count from the control broadcast)
        n = data["returns"].shape[0]
        if schema is None:
            schema = [(k, tuple(int(x) for x in data[k].shape[1:])) for k in sorted(data)]
            runtime.broadcast_object_list([schema], src=0, group=world_group)
        row_w = sum(int(np.prod(sh)) for _, sh in schema)
        base, extra = divmod(n, n_trainers)
        sizes = [base + (1 if t < extra else 0) for t in range(n_trainers)]
        rows_max = max(sizes)
        ctrl = torch.tensor([0, n], dtype=torch.int64, device=comm_dev)
        runtime.broadcast(ctrl, src=0, group=world_group)
        bufs = [torch.zeros(rows_max, row_w, dtype=torch.float32, device=comm_dev)]
        start = 0
        for t in range(n_trainers):
            buf = torch.zeros(rows_max, row_w, dtype=torch.float32, device=comm_dev)
            off = 0
            for k, sh in schema:
                w = int(np.prod(sh))
                buf[: sizes[t], off : off + w] = torch.as_tensor(
                    data[k][start : start + sizes[t]].reshape(sizes[t], w), device=comm_dev
                )
                off += w
            bufs.append(buf)
            start += sizes[t]
        runtime.scatter_tensor(bufs[0], bufs, src=0, group=world_group)

        # receive updated weights from the lead trainer
        runtime.broadcast(flat, src=1, group=pt_group)
        flat_to_params(flat, agent.parameters())

        # receive trainer metrics
        payload: List[Any] = [None]
        runtime.broadcast_object_list(payload, src=1, group=pt_group)
        for k, v in (payload[0] or {}).items():
            aggregator.update(k, v)

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
            # the player holds the freshest synced weights; the lead trainer
            # ships its optimizer state over the player<->trainer group so the
            # checkpoint is complete (resume restores it on every trainer)
            last_checkpoint = policy_step
            runtime.broadcast_object_list(["__send_opt__"], src=0, group=pt_group)
            opt_payload: List[Any] = [None]
            runtime.broadcast_object_list(opt_payload, src=1, group=pt_group)
            ckpt_path = os.path.join(log_dir, "checkpoint", f"ckpt_{policy_step}_0.ckpt")
            os.makedirs(os.path.dirname(ckpt_path), exist_ok=True)
            torch.save(
                {"agent": agent.state_dict(), "optimizer": opt_payload[0], "policy_step": policy_step},
                ckpt_path,
                pickle_protocol=4,
            )
        else:
            runtime.broadcast_object_list(["__noop__"], src=0, group=pt_group)

    # shutdown sentinel to the trainers (reference :344): control flag 1
    if schema is None:
        schema = []
        runtime.broadcast_object_list([schema], src=0, group=world_group)
    runtime.broadcast(torch.tensor([1, 0], dtype=torch.int64, device=comm_dev), src=0, group=world_group)
    envs.close()
    if cfg.algo.run_test:
        reward = test(ppo_player, make_env(cfg, cfg.seed, 0), cfg, log_dir, device)
        runtime.log_dict({"Test/cumulative_reward": reward}, policy_step)
    if logger is not None:
        logger.close()


def trainer(runtime: Runtime, cfg: Any, world_group, pt_group, opt_group) -> None:
    device = runtime.device
    rank = runtime.global_rank

    # build env spaces without running envs (dummy instance)
    probe = make_env(cfg, cfg.seed, 0)()
    obs_space = probe.observation_space
    action_space = probe.action_space
    probe.close()

    agent = PPOAgent(obs_space, action_space, cfg.algo).to(device)
    resume_opt = None
    if cfg.checkpoint.resume_from and rank == 1:
        _st = torch.load(cfg.checkpoint.resume_from, map_location="cpu", weights_only=False)
        agent.load_state_dict(_st["agent"])
        resume_opt = _st.get("optimizer")
    gs = GradSync(agent, bucket_cap_mb=runtime.bucket_cap_mb, process_group=opt_group)
    gs.broadcast_params(src=1)
    agent._grad_sync = gs  # noqa: SLF001
    runtime._synced_modules.append(agent)
    optimizer = make_optimizer(agent.parameters(), cfg.algo.optimizer)
    if cfg.checkpoint.resume_from:
        # rank 1 saved its optimizer state; every trainer resumes from it
        payload: List[Any] = [resume_opt]
        runtime.broadcast_object_list(payload, src=1, group=opt_group)
        if payload[0] is not None:
            optimizer.load_state_dict(payload[0])

    flat = params_to_flat(agent.parameters()).detach()
    if rank == 1:
        runtime.broadcast(flat, src=1, group=pt_group)

    update = 0
    comm_dev = device if runtime.backend == "nccl" else torch.device("cpu")
    schema = None
    n_trainers = runtime.world_size - 1
    while True:
        if schema is None:
            payload: List[Any] = [None]
            runtime.broadcast_object_list(payload, src=0, group=world_group)
            schema = payload[0]
        ctrl = torch.zeros(2, dtype=torch.int64, device=comm_dev)
        runtime.broadcast(ctrl, src=0, group=world_group)
        if int(ctrl[0].item()) == 1:  # shutdown (reference :463)
            return
        n = int(ctrl[1].item())
        base, extra = divmod(n, n_trainers)
        sizes = [base + (1 if t < extra else 0) for t in range(n_trainers)]
        my_rows = sizes[rank - 1]
        rows_max = max(sizes)
        row_w = sum(int(np.prod(sh)) for _, sh in schema)
        update += 1
        buf = torch.empty(rows_max, row_w, dtype=torch.float32, device=comm_dev)
        runtime.scatter_tensor(buf, None, src=0, group=world_group)
        if buf.device != device:
            buf = buf.to(device)
        data = {}
        off = 0
        for k, sh in schema:
            w = int(np.prod(sh))
            data[k] = buf[:my_rows, off : off + w].reshape(my_rows, *sh)
            off += w
        obs = {k[len("obs_") :]: data[k] for k in data if k.startswith("obs_")}
        if not agent.is_continuous:
            data["actions"] = data["actions"].long()

        losses = {"policy": 0.0, "value": 0.0, "entropy": 0.0}
        n_batches = 0
        idxs = np.arange(my_rows)
        bs = cfg.algo.per_rank_batch_size
        rounds_per_epoch = max(1, int(np.ceil(rows_max / bs)))
        for _ in range(cfg.algo.update_epochs):
            np.random.shuffle(idxs)
            for rnd in range(rounds_per_epoch):
                start = rnd * bs
                if start >= my_rows:
                    # Join-equivalent: this trainer is out of rows; shadow the
                    # peers' bucket all-reduces with zero grads and apply the
                    # averaged step so replicas stay identical
                    gs.sync_zero()
                    if cfg.algo.max_grad_norm and cfg.algo.max_grad_norm > 0:
                        runtime.clip_gradients(agent, optimizer, max_norm=cfg.algo.max_grad_norm)
                    optimizer.step()
                    continue
                sel = idxs[start : start + bs]
                batch_obs = {k: v[sel] for k, v in obs.items()}
                adv = data["advantages"][sel]
                if cfg.algo.normalize_advantages and adv.numel() > 1:
                    adv = (adv - adv.mean()) / (adv.std() + 1e-8)
                _, logprobs, entropy, new_values = agent(batch_obs, data["actions"][sel])
                pg, vl, ent = ppo_losses(
                    logprobs, data["logprobs"][sel], adv, new_values, data["values"][sel],
                    data["returns"][sel], entropy, cfg.algo.clip_coef, cfg.algo.clip_vloss,
                    cfg.algo.loss_reduction,
                )
                loss = pg + cfg.algo.vf_coef * vl + cfg.algo.ent_coef * ent
                optimizer.zero_grad(set_to_none=True)
                runtime.backward(loss)
                if cfg.algo.max_grad_norm and cfg.algo.max_grad_norm > 0:
                    runtime.clip_gradients(agent, optimizer, max_norm=cfg.algo.max_grad_norm)
                optimizer.step()
                losses["policy"] += float(pg.detach().cpu())
                losses["value"] += float(vl.detach().cpu())
                losses["entropy"] += float(ent.detach().cpu())
                n_batches += 1

        if rank == 1:
            flat = params_to_flat(agent.parameters()).detach()
            runtime.broadcast(flat, src=1, group=pt_group)
            metrics = {
                "Loss/policy_loss": losses["policy"] / max(n_batches, 1),
                "Loss/value_loss": losses["value"] / max(n_batches, 1),
                "Loss/entropy_loss": losses["entropy"] / max(n_batches, 1),
            }
            runtime.broadcast_object_list([metrics], src=1, group=pt_group)
            # checkpoint control: the player either requests the optimizer
            # state for a complete checkpoint or sends a no-op
            ctrl: List[Any] = [None]
            runtime.broadcast_object_list(ctrl, src=0, group=pt_group)
            if ctrl[0] == "__send_opt__":
                opt_sd = optimizer.state_dict()
                opt_sd = {
                    "state": {
                        k: {kk: (vv.cpu() if torch.is_tensor(vv) else vv) for kk, vv in v.items()}
                        for k, v in opt_sd["state"].items()
                    },
                    "param_groups": opt_sd["param_groups"],
                }
                runtime.broadcast_object_list([opt_sd], src=1, group=pt_group)


@register_algorithm(name="ppo_decoupled", decoupled=True)
def main(runtime: Runtime, cfg: Any) -> None:
    if runtime.world_size < 2:
        raise RuntimeError("ppo_decoupled needs at least 2 processes (1 player + >=1 trainer)")
    import torch.distributed as dist

    world_group = dist.group.WORLD
    pt_group = runtime.new_group([0, 1])
    opt_group = runtime.new_group(list(range(1, runtime.world_size)))
    if runtime.global_rank == 0:
        player(runtime, cfg, world_group, pt_group)
    else:
        trainer(runtime, cfg, world_group, pt_group, opt_group)
