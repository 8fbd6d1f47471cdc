"""Benchmark entrypoint (driver contract).

Measures the flagship training configuration — Dreamer-V3 Atari-100K
(S model: dense 512, GRU 512, CNN mult 32, batch 16 x seq 64, replay_ratio 1)
per BASELINE.json — on synthetic Atari-shaped data (no ALE in the image) with
random-init weights.  One "step" is one training iteration of the real
algorithm loop: one vectorized env interaction (policy forward + env step +
buffer add) plus the replay-ratio-driven gradient step (world model + actor +
critic update).  The reported metric is whole-job env-frames/sec
(policy steps x action_repeat x world_size / wall), the reference's headline
accounting for "Atari MsPacman 100K frames in 14h on 1x RTX 3080"
 = 400k env frames / 50400 s = 7.94 env-frames/s.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W]
For N>1 the driver launches this under torch.distributed.run (one rank/GPU,
RCCL over xGMI); each rank runs an identical replica (weak scaling, DP).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))

import numpy as np
import torch

BASELINE_ENV_FRAMES_PER_SEC = 400_000 / (14 * 3600)  # RTX 3080, BASELINE.md row 2


def _build(cfg_overrides, device_type: str):
    from sheeprl_amd.config import compose

    overrides = [
        "exp=dreamer_v3_100k_ms_pacman",
        f"runtime.accelerator={device_type}",
        "runtime.precision=" + ("bf16" if device_type == "cuda" else "fp32"),
        "metric.log_level=0",
        "metric.disable_timer=True",
        "checkpoint.every=0",
        "checkpoint.save_last=False",
        "buffer.size=4096",
        "algo.run_test=False",
        "env.sync_env=True",
    ] + list(cfg_overrides)
    return compose(overrides)


def _setup(cfg, rank: int, world_size: int):
    """Build env, agent, buffer, optimizers — the same components the real
    training loop uses (sheeprl_amd/algos/dreamer_v3/dreamer_v3.py)."""
    import torch.nn.functional as F

    from sheeprl_amd.algos.dreamer_v3.agent import build_agent
    from sheeprl_amd.algos.dreamer_v3.utils import Moments, prepare_obs
    from sheeprl_amd.data import EnvIndependentReplayBuffer, SequentialReplayBuffer
    from sheeprl_amd.envs import vectorize_env
    from sheeprl_amd.optim import FusedAdam
    from sheeprl_amd.parallel import Runtime
    from sheeprl_amd.utils.utils import seed_everything

    seed_everything(cfg.seed + rank)
    torch.backends.cudnn.benchmark = True  # MIOpen find mode: avoid naive-conv fallback
    torch.set_float32_matmul_precision("high")
    runtime = Runtime(
        devices=world_size,
        accelerator=cfg.runtime.accelerator,
        precision=cfg.runtime.precision,
    )
    runtime.world_size = world_size
    runtime.global_rank = rank
    runtime.local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if world_size > 1:
        runtime._init_process_group(init_method="env://")
    else:
        runtime._setup_device()

    envs = vectorize_env(cfg, cfg.seed, rank)
    obs_space = envs.single_observation_space
    action_space = envs.single_action_space
    actions_dim = [action_space.n]

    world_model, actor, critic, target_critic, player = build_agent(
        runtime, actions_dim, False, cfg, obs_space
    )
    world_optimizer = FusedAdam(world_model.parameters(), lr=cfg.algo.world_model.optimizer.lr,
                                eps=cfg.algo.world_model.optimizer.eps)
    actor_optimizer = FusedAdam(actor.parameters(), lr=cfg.algo.actor.optimizer.lr, eps=cfg.algo.actor.optimizer.eps)
    critic_optimizer = FusedAdam(critic.parameters(), lr=cfg.algo.critic.optimizer.lr,
                                 eps=cfg.algo.critic.optimizer.eps)
    moments = Moments(
        cfg.algo.actor.moments.decay,
        cfg.algo.actor.moments.max,
        cfg.algo.actor.moments.percentile.low,
        cfg.algo.actor.moments.percentile.high,
    ).to(runtime.device)

    rb = EnvIndependentReplayBuffer(
        int(cfg.buffer.size),
        n_envs=cfg.env.num_envs,
        obs_keys=list(cfg.algo.cnn_keys.encoder),
        buffer_cls=SequentialReplayBuffer,
        # pinned host ring: device-readable zero-copy for the HIP replay
        # gather
        pinned=runtime.use_cuda,
    )
    return runtime, envs, (world_model, actor, critic, target_critic, player), (
        world_optimizer, actor_optimizer, critic_optimizer), moments, rb


def _prefill(cfg, envs, rb, n_steps: int):
    """Fill the replay buffer with random-policy synthetic data (untimed)."""
    num_envs = cfg.env.num_envs
    n_act = envs.single_action_space.n
    obs, _ = envs.reset(seed=cfg.seed)
    step_data = {}
    cnn_keys = list(cfg.algo.cnn_keys.encoder)
    for k in cnn_keys:
        step_data[k] = np.asarray(obs[k])[None]
    step_data["rewards"] = np.zeros((1, num_envs, 1), np.float32)
    step_data["terminated"] = np.zeros((1, num_envs, 1), np.float32)
    step_data["truncated"] = np.zeros((1, num_envs, 1), np.float32)
    step_data["is_first"] = np.ones_like(step_data["terminated"])
    rng = np.random.default_rng(0)
    for _ in range(n_steps):
        a = rng.integers(0, n_act, size=num_envs)
        onehot = np.eye(n_act, dtype=np.float32)[a]
        step_data["actions"] = onehot[None]
        rb.add(step_data)
        obs, rewards, term, trunc, _ = envs.step(a)
        for k in cnn_keys:
            step_data[k] = np.asarray(obs[k])[None]
        step_data["rewards"] = np.asarray(rewards, np.float32).reshape(1, num_envs, 1)
        step_data["terminated"] = np.asarray(term, np.float32).reshape(1, num_envs, 1)
        step_data["truncated"] = np.asarray(trunc, np.float32).reshape(1, num_envs, 1)
        step_data["is_first"] = np.zeros_like(step_data["terminated"])
    return obs, step_data


def run_bench(n_gpus: int, steps: int, warmup: int, overrides=()) -> dict:
    import torch.distributed as dist

    from sheeprl_amd.algos.dreamer_v3.dreamer_v3 import train
    from sheeprl_amd.algos.dreamer_v3.utils import prepare_obs
    from sheeprl_amd.utils.metric import MetricAggregator

    rank = int(os.environ.get("RANK", 0))
    world_size = int(os.environ.get("WORLD_SIZE", n_gpus))
    device_type = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = _build(overrides, device_type)
    if device_type == "cuda":
        from sheeprl_amd.ops import has_ext

        if not has_ext():
            raise RuntimeError("HIP extension _sheep_hip is not built — refusing to bench the eager fallback")
        # hipBLASLt algorithm tuning during the (untimed) warmup: the scan's
        # M=16 long-K GEMMs gain ~1 ms/step on XL from better algo picks
        # (neutral on S, measured).  Single-rank only — tuning sweeps would
        # desync multi-rank warmup; opt out with SHEEPRL_AMD_TUNABLEOP=0.
        if world_size == 1 and os.environ.get("SHEEPRL_AMD_TUNABLEOP", "1") != "0":
            try:
                import torch.cuda.tunable as tunable

                tunable.set_filename(os.environ.get("TMPDIR", "/tmp") + "/sheeprl_tunableop.csv")
                tunable.enable(True)
            except Exception as e:  # noqa: BLE001
                print(f"[bench] TunableOp unavailable ({e})", file=sys.stderr)

    runtime, envs, models, optims, moments, rb = _setup(cfg, rank, world_size)
    world_model, actor, critic, target_critic, player = models
    world_optimizer, actor_optimizer, critic_optimizer = optims
    device = runtime.device
    num_envs = cfg.env.num_envs
    actions_dim = [envs.single_action_space.n]
    n_act = actions_dim[0]
    seq_len = cfg.algo.per_rank_sequence_length

    obs, step_data = _prefill(cfg, envs, rb, n_steps=max(seq_len + 1, 80))
    player.init_states()

    aggregator = MetricAggregator({})
    MetricAggregator.disabled = True

    # pinned-host prefetch: the next minibatch is gathered + staged on a side
    # stream while the GPU executes the current gradient step
    from sheeprl_amd.data.prefetch import DevicePrefetcher

    def _sample_host():
        s = rb.sample_tensors(
            cfg.algo.per_rank_batch_size, sequence_length=seq_len, n_samples=1, device="cpu"
        )
        return {k: v[0] for k, v in s.items()}

    # measured on MI355X: the async graph replay already hides the ~1.5 ms
    # host sample, and the prefetch thread costs ~10% (GIL + pinned memcpy
    # contention with the env step) — so default OFF; flip on for slow hosts
    use_prefetch = os.environ.get("SHEEPRL_AMD_PREFETCH", "0") == "1"
    prefetcher = DevicePrefetcher(_sample_host, device, depth=1) if use_prefetch else None

    # device-side replay gather: HIP kernel pulls sequence windows from the
    # pinned ring straight into HBM on a side stream, one batch ahead
    gatherer = None
    if device_type == "cuda" and os.environ.get("SHEEPRL_AMD_DEVICE_GATHER", "1") == "1" and prefetcher is None:
        from sheeprl_amd.data.gather import DeviceReplayGather

        try:
            gatherer = DeviceReplayGather(rb, cfg.algo.per_rank_batch_size, seq_len, device)
        except Exception as e:  # noqa: BLE001
            print(f"[bench] device gather unavailable ({e}); host sampling", file=sys.stderr)

    class GraphedPlayer:
        """hipGraph-captured env-interaction forward: the per-step player
        chain (encoder -> GRU cell -> representation -> actor) is ~30 small
        kernels on a latency chain (~1.4 ms/step measured); one replay brings
        it to a single graph launch.  States live in fixed buffers; episode
        resets write them in place between replays."""

        def __init__(self, player, torch_obs):
            self.player = player
            self.obs_keys = list(torch_obs.keys())
            self.static_obs = {k: v.clone() for k, v in torch_obs.items()}
            self.h = player.recurrent_state.clone()
            self.z = player.stochastic_state.clone()
            self.a = player.actions.clone()
            self.out = None
            rssm, actor_m, enc = player.rssm, player.actor, player.encoder

            @torch.inference_mode()
            def fwd():
                emb = enc(self.static_obs)
                rec = rssm.recurrent_model(torch.cat((self.z, self.a), -1), self.h)
                _, stoch = rssm._representation(rec, emb)
                z_new = stoch.view(*stoch.shape[:-2], -1)
                acts, _ = actor_m(torch.cat((z_new, rec), -1))
                a_new = torch.cat(acts, -1).to(z_new.dtype)
                self.h.copy_(rec)
                self.z.copy_(z_new)
                self.a.copy_(a_new)
                return a_new

            stream = torch.cuda.Stream()
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                for _ in range(2):
                    out = fwd()
            torch.cuda.current_stream().wait_stream(stream)
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                self.out = fwd()

        def get_actions(self, torch_obs):
            for k in self.obs_keys:
                self.static_obs[k].copy_(torch_obs[k])
            self.graph.replay()
            return [self.out[0]]

        def init_states(self, reset_envs):
            rec, stoch = self.player.rssm.get_initial_states((1, len(reset_envs)))
            self.a[:, reset_envs] = 0.0
            self.h[:, reset_envs] = rec.to(self.h.dtype)
            self.z[:, reset_envs] = stoch.reshape(1, len(reset_envs), -1).to(self.z.dtype)

    def train_fn(batch):
        train(
            runtime, world_model, actor, critic, target_critic,
            world_optimizer, actor_optimizer, critic_optimizer,
            batch, aggregator, cfg, False, actions_dim, moments,
        )

    # hipGraph-capture the whole gradient step (fwd+bwd+optimizers): the
    # T=64 RSSM scan + imagination is ~20k tiny kernels, host-bound in eager.
    graphed = None
    if device_type == "cuda" and os.environ.get("SHEEPRL_AMD_NO_GRAPHS", "0") != "1":
        from sheeprl_amd.parallel.graphs import CUDAGraphStep

        example = {
            k: v[0] for k, v in rb.sample_tensors(
                cfg.algo.per_rank_batch_size, sequence_length=seq_len, n_samples=1, device=device
            ).items()
        }
        try:
            graphed = CUDAGraphStep(train_fn, example, warmup=3)
            if rank == 0:
                print("[bench] train step captured in a hipGraph", file=sys.stderr)
        except Exception as e:  # noqa: BLE001
            graphed = None
            if rank == 0:
                print(f"[bench] hipGraph capture failed ({e}); running eager", file=sys.stderr)

    # graph-capture the player forward too (opt-out: SHEEPRL_AMD_NO_GRAPHS)
    gplayer = None
    if graphed is not None and not player.actor.is_continuous and len(actions_dim) == 1:
        try:
            with torch.no_grad():
                _tobs = prepare_obs(runtime, obs, cnn_keys=list(cfg.algo.cnn_keys.encoder), num_envs=num_envs)
                gplayer = GraphedPlayer(player, _tobs)
            if rank == 0:
                print("[bench] player forward captured in a hipGraph", file=sys.stderr)
        except Exception as e:  # noqa: BLE001
            gplayer = None
            if rank == 0:
                print(f"[bench] player graph capture failed ({e}); eager player", file=sys.stderr)

    def one_iter(obs, step_data):
        # --- env interaction (the real policy forward + env step + buffer add)
        with torch.inference_mode():
            torch_obs = prepare_obs(runtime, obs, cnn_keys=list(cfg.algo.cnn_keys.encoder), num_envs=num_envs)
            acts = (gplayer or player).get_actions(torch_obs)
            actions = torch.cat(acts, -1).view(num_envs, -1).float().cpu().numpy()
            real_actions = torch.stack([a.argmax(dim=-1) for a in acts], dim=-1).view(num_envs).cpu().numpy()
        step_data["actions"] = actions[None]
        if prefetcher is not None:
            with prefetcher.lock:
                rb.add(step_data)
        else:
            if gatherer is not None:
                # a pending gather may read the ring row this add overwrites
                gatherer.sync()
            rb.add(step_data)
        obs, rewards, term, trunc, infos = envs.step(real_actions)
        for k in cfg.algo.cnn_keys.encoder:
            step_data[k] = np.asarray(obs[k])[None]
        step_data["rewards"] = np.asarray(rewards, np.float32).reshape(1, num_envs, 1)
        step_data["terminated"] = np.asarray(term, np.float32).reshape(1, num_envs, 1)
        step_data["truncated"] = np.asarray(trunc, np.float32).reshape(1, num_envs, 1)
        step_data["is_first"] = np.zeros_like(step_data["terminated"])
        dones = np.logical_or(term, trunc)
        if dones.any():
            # player states were produced under inference_mode; resetting them
            # in place must happen under it too
            with torch.inference_mode():
                (gplayer or player).init_states(np.nonzero(dones)[0].tolist())
        # --- one gradient step (replay_ratio=1 at num_envs=1)
        from sheeprl_amd import ops as _ops

        _ops.ema_update_(list(target_critic.parameters()), list(critic.parameters()), cfg.algo.critic.tau)
        if prefetcher is not None:
            batch = prefetcher.next()
        elif gatherer is not None:
            batch = {k: v[0] for k, v in gatherer.next().items()}
        else:
            s = rb.sample_tensors(cfg.algo.per_rank_batch_size, sequence_length=seq_len, n_samples=1, device=device)
            batch = {k: v[0] for k, v in s.items()}
        if graphed is not None:
            graphed(batch)
        else:
            train_fn(batch)
        return obs

    # warmup
    for _ in range(warmup):
        obs = one_iter(obs, step_data)

    if os.environ.get("SHEEPRL_AMD_PHASE_TIMING"):
        # coarse phase split over a few synchronized steps
        phases = {"env": 0.0, "sample": 0.0, "train": 0.0}
        for _ in range(5):
            torch.cuda.synchronize()
            t = time.perf_counter()
            with torch.inference_mode():
                torch_obs = prepare_obs(runtime, obs, cnn_keys=list(cfg.algo.cnn_keys.encoder), num_envs=num_envs)
                acts = player.get_actions(torch_obs)
                real_actions = torch.stack([a.argmax(dim=-1) for a in acts], dim=-1).view(num_envs).cpu().numpy()
            obs2, rewards, term, trunc, infos = envs.step(real_actions)
            torch.cuda.synchronize()
            phases["env"] += time.perf_counter() - t
            t = time.perf_counter()
            if prefetcher is not None:
                batch = prefetcher.next()
            else:
                _s = rb.sample_tensors(cfg.algo.per_rank_batch_size, sequence_length=seq_len, n_samples=1, device=device)
                batch = {k: v[0] for k, v in _s.items()}
            torch.cuda.synchronize()
            phases["sample"] += time.perf_counter() - t
            t = time.perf_counter()
            if graphed is not None:
                graphed(batch)
            else:
                train_fn(batch)
            torch.cuda.synchronize()
            phases["train"] += time.perf_counter() - t
            obs = obs2
        print({k: round(v / 5 * 1000, 2) for k, v in phases.items()}, file=sys.stderr)

    if os.environ.get("SHEEPRL_AMD_TORCH_PROFILE"):
        from torch.profiler import ProfilerActivity, profile

        with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
            for _ in range(2):
                obs = one_iter(obs, step_data)
        out = os.environ["SHEEPRL_AMD_TORCH_PROFILE"]
        with open(out, "w") as f:
            f.write(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=40))
            f.write("\n\n")
            f.write(prof.key_averages().table(sort_by="self_cpu_time_total", row_limit=40))

    if runtime.is_distributed:
        dist.barrier()
    if device_type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        obs = one_iter(obs, step_data)
    if device_type == "cuda":
        torch.cuda.synchronize()
    if runtime.is_distributed:
        dist.barrier()
    elapsed = time.perf_counter() - t0
    if runtime.is_distributed:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device if runtime.backend == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if prefetcher is not None:
        prefetcher.close()
    policy_steps = steps * num_envs * world_size
    env_frames_per_sec = policy_steps * cfg.env.action_repeat / elapsed
    result = {
        "metric": "env_frames_per_sec",
        "value": round(env_frames_per_sec, 3),
        "unit": "frames/s",
        "n_gpus": world_size,
        "steps": steps,
        "warmup": warmup,
        "ms_per_step": round(elapsed / steps * 1000, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": round(env_frames_per_sec / (BASELINE_ENV_FRAMES_PER_SEC * 1), 3),
        "dtype": "bf16" if device_type == "cuda" else "fp32",
        "data": "synthetic",
        "config": {
            "model": {256: "dreamer_v3_XS", 512: "dreamer_v3_S", 1024: "dreamer_v3_M",
                      2048: "dreamer_v3_L", 4096: "dreamer_v3_XL"}.get(
                cfg.algo.world_model.recurrent_model.recurrent_state_size, "dreamer_v3_custom"),
            "global_batch": cfg.algo.per_rank_batch_size * world_size,
            "seq_len": seq_len,
            "parallelism": f"dp{world_size}",
            "benchmark": "DreamerV3 Atari-100K (MsPacman shape, synthetic env)",
            "grad_steps_per_policy_step": cfg.algo.replay_ratio,
            "action_repeat": cfg.env.action_repeat,
        },
    }
    envs.close()
    return result


def run_bench_sac(n_gpus: int, steps: int, warmup: int, overrides=()) -> dict:
    """SAC on a HalfCheetah-shaped synthetic env (17-dim obs, 6-dim actions),
    bf16 on GPU — BASELINE.json config #2.  One step = one vectorized env
    interaction + the replay-ratio-driven gradient step (batch 256), with the
    whole gradient step captured in a hipGraph."""
    import torch.distributed as dist

    from sheeprl_amd.algos.sac.agent import build_agent as build_sac
    from sheeprl_amd.algos.sac.sac import train as sac_train
    from sheeprl_amd.config import compose
    from sheeprl_amd.data import ReplayBuffer
    from sheeprl_amd.envs import vectorize_env
    from sheeprl_amd.optim import FusedAdam
    from sheeprl_amd.parallel import Runtime
    from sheeprl_amd.utils.metric import MetricAggregator
    from sheeprl_amd.utils.utils import seed_everything

    rank = int(os.environ.get("RANK", 0))
    world_size = int(os.environ.get("WORLD_SIZE", n_gpus))
    device_type = "cuda" if torch.cuda.is_available() else "cpu"
    if device_type == "cuda":
        from sheeprl_amd.ops import has_ext

        if not has_ext():
            raise RuntimeError("HIP extension _sheep_hip is not built — refusing to bench the eager fallback")
    cfg = compose([
        "exp=sac",
        "env=dummy",
        "env.id=dummy_continuous",
        "env.num_envs=4",
        "env.max_episode_steps=1000",
        "algo.mlp_keys.encoder=[state]",
        f"runtime.accelerator={device_type}",
        "runtime.precision=" + ("bf16" if device_type == "cuda" else "fp32"),
        "metric.log_level=0",
        "metric.disable_timer=True",
        "checkpoint.every=0",
        "checkpoint.save_last=False",
        "buffer.size=65536",
        "buffer.memmap=False",
        "algo.run_test=False",
    ] + list(overrides))
    cfg.env.wrapper_kwargs = {"state_dim": 17, "action_dim": 6, "n_steps": 1000}
    seed_everything(cfg.seed + rank)
    runtime = Runtime(devices=world_size, accelerator=cfg.runtime.accelerator, precision=cfg.runtime.precision)
    runtime.world_size = world_size
    runtime.global_rank = rank
    runtime.local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if world_size > 1:
        runtime._init_process_group(init_method="env://")
    else:
        runtime._setup_device()
    device = runtime.device
    num_envs = cfg.env.num_envs

    envs = vectorize_env(cfg, cfg.seed, rank)
    obs_space = envs.single_observation_space
    action_space = envs.single_action_space
    agent, player = build_sac(runtime, cfg, obs_space, action_space, None)
    qf_optimizer = FusedAdam(agent.qfs.parameters(), lr=cfg.algo.critic.optimizer.lr)
    actor_optimizer = FusedAdam(agent.actor.parameters(), lr=cfg.algo.actor.optimizer.lr)
    alpha_optimizer = FusedAdam([agent.log_alpha], lr=cfg.algo.alpha.optimizer.lr)
    rb = ReplayBuffer(int(cfg.buffer.size), num_envs, obs_keys=("obs",), memmap=False)

    obs_np, _ = envs.reset(seed=cfg.seed)
    obs = np.asarray(obs_np["state"], np.float32).reshape(num_envs, -1)
    rng = np.random.default_rng(0)
    bs = cfg.algo.per_rank_batch_size

    def env_step(actions):
        nonlocal obs
        nxt, rewards, terms, truncs, _ = envs.step(actions)
        nxt_flat = np.asarray(nxt["state"], np.float32).reshape(num_envs, -1)
        rb.add({
            "obs": obs[None],
            "next_obs": nxt_flat[None],
            "actions": actions[None].astype(np.float32),
            "rewards": np.asarray(rewards, np.float32).reshape(1, num_envs, 1),
            "dones": np.logical_or(terms, truncs).astype(np.float32).reshape(1, num_envs, 1),
        })
        obs = nxt_flat

    # random prefill (untimed)
    for _ in range(max(bs // num_envs + 2, 70)):
        env_step(rng.uniform(-1, 1, size=(num_envs, 6)).astype(np.float32))

    aggregator = MetricAggregator({})
    MetricAggregator.disabled = True
    update = [0]

    def train_fn(batch):
        update[0] += 1
        sac_train(runtime, agent, actor_optimizer, qf_optimizer, alpha_optimizer,
                  batch, None, update[0], cfg, num_envs * world_size)

    pdt = runtime.param_dtype

    def cast_batch(batch):
        # model runs in param_dtype (bf16 on GPU); rewards/dones stay fp32
        for k in ("obs", "next_obs", "actions"):
            batch[k] = batch[k].to(pdt)
        return batch

    graphed = None
    if device_type == "cuda" and os.environ.get("SHEEPRL_AMD_NO_GRAPHS", "0") != "1":
        from sheeprl_amd.parallel.graphs import CUDAGraphStep

        example = cast_batch({k: v[0] for k, v in rb.sample_tensors(bs, n_samples=1, device=device).items()})
        try:
            graphed = CUDAGraphStep(train_fn, example, warmup=3)
            if rank == 0:
                print("[bench] SAC train step captured in a hipGraph", file=sys.stderr)
        except Exception as e:  # noqa: BLE001
            graphed = None
            if rank == 0:
                print(f"[bench] hipGraph capture failed ({e}); running eager", file=sys.stderr)

    def one_iter():
        with torch.no_grad():
            t_obs = torch.as_tensor(obs, device=device, dtype=torch.float32).to(pdt)
            actions = player.get_actions(t_obs).float().cpu().numpy()
        env_step(actions)
        batch = cast_batch({k: v[0] for k, v in rb.sample_tensors(bs, n_samples=1, device=device).items()})
        if graphed is not None:
            graphed(batch)
        else:
            train_fn(batch)

    for _ in range(warmup):
        one_iter()
    if runtime.is_distributed:
        dist.barrier()
    if device_type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        one_iter()
    if device_type == "cuda":
        torch.cuda.synchronize()
    if runtime.is_distributed:
        dist.barrier()
    elapsed = time.perf_counter() - t0
    if runtime.is_distributed:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device if runtime.backend == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    envs.close()
    fps = steps * num_envs * world_size / elapsed
    return {
        "metric": "env_frames_per_sec",
        "value": round(fps, 3),
        "unit": "frames/s",
        "n_gpus": world_size,
        "steps": steps,
        "warmup": warmup,
        "ms_per_step": round(elapsed / steps * 1000, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16" if device_type == "cuda" else "fp32",
        "data": "synthetic",
        "config": {
            "model": "sac",
            "global_batch": bs * world_size,
            "seq_len": 1,
            "parallelism": f"dp{world_size}",
            "benchmark": "SAC HalfCheetah-shape (17-dim obs, 6-dim act, synthetic env)",
            "grad_steps_per_policy_step": cfg.algo.replay_ratio,
            "action_repeat": 1,
        },
    }


def run_bench_ppo(n_gpus: int, steps: int, warmup: int, overrides=()) -> dict:
    """PPO on the synthetic Atari-shaped env (3x64x64 uint8, 9 actions),
    bf16 on GPU.  One step = one full PPO iteration: rollout_steps env
    interactions per env + GAE + update_epochs x minibatch SGD, the
    minibatch step captured in a hipGraph (minibatch shapes are constant).
    BASELINE.json names PPO as part of the headline metric; the reference's
    published PPO row is the CPU wall-clock benchmark (benchmarks/RESULTS.md)."""
    import torch.distributed as dist

    from sheeprl_amd.algos.ppo.agent import build_agent as build_ppo
    from sheeprl_amd.algos.ppo.loss import entropy_loss, policy_loss, value_loss
    from sheeprl_amd.algos.ppo.utils import prepare_obs
    from sheeprl_amd.config import compose
    from sheeprl_amd.envs import vectorize_env
    from sheeprl_amd.optim import make_optimizer
    from sheeprl_amd.parallel import Runtime
    from sheeprl_amd.ops import gae as compute_gae
    from sheeprl_amd.utils.utils import seed_everything

    rank = int(os.environ.get("RANK", 0))
    world_size = int(os.environ.get("WORLD_SIZE", n_gpus))
    device_type = "cuda" if torch.cuda.is_available() else "cpu"
    if device_type == "cuda":
        from sheeprl_amd.ops import has_ext

        if not has_ext():
            raise RuntimeError("HIP extension _sheep_hip is not built — refusing to bench the eager fallback")
    cfg = compose([
        "exp=ppo",
        "env=synthetic_atari",
        "env.num_envs=8",
        "algo.cnn_keys.encoder=[rgb]",
        "algo.mlp_keys.encoder=[]",
        "algo.rollout_steps=128",
        "algo.per_rank_batch_size=256",
        "algo.update_epochs=4",
        f"runtime.accelerator={device_type}",
        "runtime.precision=" + ("bf16" if device_type == "cuda" else "fp32"),
        "metric.log_level=0",
        "metric.disable_timer=True",
        "checkpoint.every=0",
        "checkpoint.save_last=False",
        "algo.run_test=False",
        "env.sync_env=True",
    ] + list(overrides))
    seed_everything(cfg.seed + rank)
    runtime = Runtime(devices=world_size, accelerator=cfg.runtime.accelerator, precision=cfg.runtime.precision)
    runtime.world_size = world_size
    runtime.global_rank = rank
    runtime.local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if world_size > 1:
        runtime._init_process_group(init_method="env://")
    else:
        runtime._setup_device()
    device = runtime.device
    num_envs = cfg.env.num_envs
    T = cfg.algo.rollout_steps
    bs = cfg.algo.per_rank_batch_size
    rows = T * num_envs

    envs = vectorize_env(cfg, cfg.seed, rank)
    obs_space = envs.single_observation_space
    action_space = envs.single_action_space
    agent, player = build_ppo(runtime, obs_space, action_space, cfg, None)
    optimizer = make_optimizer(agent.parameters(), cfg.algo.optimizer)

    # device-side rollout storage (obs stay uint8; encoder normalizes on GPU)
    store = {
        "obs_rgb": torch.empty(T, num_envs, *obs_space["rgb"].shape, device=device, dtype=torch.uint8),
        "actions": torch.empty(T, num_envs, 1, device=device, dtype=torch.float32),
        "logprobs": torch.empty(T, num_envs, 1, device=device, dtype=torch.float32),
        "values": torch.empty(T, num_envs, 1, device=device, dtype=torch.float32),
        "rewards": torch.empty(T, num_envs, 1, device=device, dtype=torch.float32),
        "dones": torch.empty(T, num_envs, 1, device=device, dtype=torch.float32),
    }
    obs, _ = envs.reset(seed=cfg.seed + rank * num_envs)

    normalize = cfg.algo.normalize_advantages

    def mb_train(batch):
        obs_b = {"rgb": batch["obs_rgb"]}
        adv = batch["advantages"]
        if normalize:
            adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        _, logprobs, entropy, new_values = agent(obs_b, batch["actions"])
        pg_loss = policy_loss(logprobs, batch["logprobs"], adv, cfg.algo.clip_coef, cfg.algo.loss_reduction)
        # bf16-true agent emits bf16 values; targets are stored fp32
        v_loss = value_loss(new_values.float(), batch["values"], batch["returns"], cfg.algo.clip_coef,
                            cfg.algo.clip_vloss, cfg.algo.loss_reduction)
        ent_loss = entropy_loss(entropy, cfg.algo.loss_reduction)
        loss = pg_loss + cfg.algo.vf_coef * v_loss + cfg.algo.ent_coef * ent_loss
        optimizer.zero_grad(set_to_none=True)
        runtime.backward(loss)
        if cfg.algo.max_grad_norm and cfg.algo.max_grad_norm > 0:
            runtime.clip_gradients(agent, optimizer, max_norm=cfg.algo.max_grad_norm)
        optimizer.step()

    # hipGraph-captured player forward: obs copies into a static uint8
    # buffer, one replay runs encoder+actor+critic+sample (philox-only RNG),
    # and the actions/logprobs/values are read from static outputs.  The
    # eager chain measured ~0.9 ms x 128 rollout steps per iteration.
    gplayer = None
    if device_type == "cuda" and os.environ.get("SHEEPRL_AMD_NO_GRAPHS", "0") != "1":
        class _GraphedPPOPlayer:
            def __init__(self, t_obs):
                self.obs_buf = {k: v.clone() for k, v in t_obs.items()}
                for _ in range(3):
                    player.get_actions(self.obs_buf)
                torch.cuda.synchronize()
                self.graph = torch.cuda.CUDAGraph()
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    player.get_actions(self.obs_buf)
                torch.cuda.current_stream().wait_stream(s)
                with torch.cuda.graph(self.graph):
                    self.out = player.get_actions(self.obs_buf)

            def get_actions(self, t_obs):
                for k, v in t_obs.items():
                    self.obs_buf[k].copy_(v)
                self.graph.replay()
                return self.out

        try:
            gplayer = _GraphedPPOPlayer(prepare_obs(obs, cfg, device))
            if rank == 0:
                print("[bench] PPO player forward captured in a hipGraph", file=sys.stderr)
        except Exception as e:  # noqa: BLE001
            gplayer = None
            if rank == 0:
                print(f"[bench] PPO player graph capture failed ({e}); eager player", file=sys.stderr)

    def rollout():
        nonlocal obs
        for t in range(T):
            t_obs = prepare_obs(obs, cfg, device)
            with torch.no_grad():
                actions, logprobs, values = (gplayer or player).get_actions(t_obs)
            env_actions = actions.cpu().numpy()[..., 0]
            next_obs, rewards, terms, truncs, _ = envs.step(env_actions)
            store["obs_rgb"][t].copy_(t_obs["rgb"])
            store["actions"][t].copy_(actions.float())
            store["logprobs"][t].copy_(logprobs.float())
            store["values"][t].copy_(values.float())
            store["rewards"][t].copy_(torch.as_tensor(rewards, dtype=torch.float32).reshape(num_envs, 1).to(device))
            store["dones"][t].copy_(torch.as_tensor(
                np.logical_or(terms, truncs).astype(np.float32)).reshape(num_envs, 1).to(device))
            obs = next_obs
        with torch.no_grad():
            next_values = player.get_values(prepare_obs(obs, cfg, device))
        returns, advantages = compute_gae(
            store["rewards"], store["values"], store["dones"].bool(), next_values,
            T, cfg.algo.gamma, cfg.algo.gae_lambda)
        data = {k: v.reshape(rows, *v.shape[2:]) for k, v in store.items()}
        data["returns"] = returns.reshape(rows, 1)
        data["advantages"] = advantages.reshape(rows, 1)
        data["actions"] = data["actions"].long()
        return data

    graphed = None
    if device_type == "cuda" and os.environ.get("SHEEPRL_AMD_NO_GRAPHS", "0") != "1":
        from sheeprl_amd.parallel.graphs import CUDAGraphStep

        data = rollout()
        example = {k: v[:bs].clone() for k, v in data.items()}
        try:
            graphed = CUDAGraphStep(mb_train, example, warmup=3)
            if rank == 0:
                print("[bench] PPO minibatch step captured in a hipGraph", file=sys.stderr)
        except Exception as e:  # noqa: BLE001
            graphed = None
            if rank == 0:
                print(f"[bench] hipGraph capture failed ({e}); running eager", file=sys.stderr)

    gen = torch.Generator(device=device)
    gen.manual_seed(cfg.seed + rank)

    def one_iter():
        data = rollout()
        for _ in range(cfg.algo.update_epochs):
            perm = torch.randperm(rows, device=device, generator=gen)
            for s in range(0, rows, bs):
                idx = perm[s:s + bs]
                batch = {k: v[idx] for k, v in data.items()}
                if graphed is not None:
                    graphed(batch)
                else:
                    mb_train(batch)

    for _ in range(warmup):
        one_iter()
    if runtime.is_distributed:
        dist.barrier()
    if device_type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        one_iter()
    if device_type == "cuda":
        torch.cuda.synchronize()
    if runtime.is_distributed:
        dist.barrier()
    elapsed = time.perf_counter() - t0
    if runtime.is_distributed:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device if runtime.backend == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    envs.close()
    fps = steps * rows * world_size / elapsed
    return {
        "metric": "env_frames_per_sec",
        "value": round(fps, 3),
        "unit": "frames/s",
        "n_gpus": world_size,
        "steps": steps,
        "warmup": warmup,
        "ms_per_step": round(elapsed / steps * 1000, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16" if device_type == "cuda" else "fp32",
        "data": "synthetic",
        "config": {
            "model": "ppo_pixel",
            "global_batch": bs * world_size,
            "seq_len": 1,
            "parallelism": f"dp{world_size}",
            "benchmark": f"PPO pixels (3x64x64 synthetic Atari shape, rollout {T} x {num_envs} envs, "
                         f"{cfg.algo.update_epochs} epochs)",
            "rollout_steps": T,
            "update_epochs": cfg.algo.update_epochs,
            "action_repeat": 1,
        },
    }


def smoke_step() -> None:
    """One tiny forward+backward of the flagship model on cuda:0 (driver
    contract: __graft_entry__.smoke)."""
    res = run_bench(
        1,
        steps=2,
        warmup=0,
        overrides=[
            "algo=dreamer_v3_XS",
            "algo.per_rank_batch_size=4",
            "algo.per_rank_sequence_length=8",
            "buffer.size=256",
        ],
    )
    print("smoke:", json.dumps(res))


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=100)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--algo", choices=["dreamer_v3", "sac", "ppo"], default="dreamer_v3",
                   help="flagship DV3 Atari-100K (default), SAC HalfCheetah-shape (BASELINE #2), "
                        "or PPO pixels (synthetic Atari shape)")
    p.add_argument("--override", action="append", default=[])
    args = p.parse_args()
    rank = int(os.environ.get("RANK", 0))
    if args.algo == "sac":
        result = run_bench_sac(args.gpus, args.steps, args.warmup, args.override)
    elif args.algo == "ppo":
        result = run_bench_ppo(args.gpus, args.steps, args.warmup, args.override)
    else:
        result = run_bench(args.gpus, args.steps, args.warmup, args.override)
    if rank == 0:
        print(json.dumps(result))


if __name__ == "__main__":
    main()
