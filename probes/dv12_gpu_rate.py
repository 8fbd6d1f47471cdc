import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
"""Graphed gradient-step rate for Dreamer-V1/V2 at their default (paper)
dims, bf16, synthetic batch — breadth evidence beside the DV3 flagship."""
import time
import numpy as np
import torch

from sheeprl_amd.config import compose
from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.utils import seed_everything


def rate(exp, capture_mod, build, n=50):
    cfg = compose([
        f"exp={exp}", "env=synthetic_atari", "runtime.accelerator=cuda",
        "runtime.precision=bf16", "metric.log_level=0", "checkpoint.every=0",
    ])
    seed_everything(cfg.seed)
    torch.backends.cudnn.benchmark = True
    runtime = Runtime(devices=1, accelerator="cuda", precision="bf16")
    runtime.world_size = 1
    runtime.global_rank = 0
    runtime.local_rank = 0
    runtime._setup_device()
    T = cfg.algo.per_rank_sequence_length
    B = cfg.algo.per_rank_batch_size
    A = 9
    dev = runtime.device
    data = {
        "rgb": torch.randint(0, 256, (T, B, 3, 64, 64), device=dev, dtype=torch.uint8),
        "actions": torch.nn.functional.one_hot(
            torch.randint(0, A, (T, B), device=dev), A).float(),
        "rewards": torch.randn(T, B, 1, device=dev) * 0.1,
        "is_first": (torch.rand(T, B, 1, device=dev) < 0.05).float(),
        "terminated": torch.zeros(T, B, 1, device=dev),
        "truncated": torch.zeros(T, B, 1, device=dev),
    }
    data["is_first"][0] = 1.0
    step = build(runtime, cfg, A, data, capture_mod)
    for _ in range(5):
        step(data)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        step(data)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / n * 1000
    print(f"{exp}: {ms:.2f} ms per graphed gradient step "
          f"(batch {B} x seq {T}, default dims, bf16)")


def build_dv2(runtime, cfg, A, data, mod):
    from sheeprl_amd.algos.dreamer_v2.agent import build_agent
    from sheeprl_amd.optim import make_optimizer

    obs_space = _space()
    wm, actor, critic, target_critic, _ = build_agent(runtime, [A], False, cfg, obs_space)
    wo = make_optimizer(wm.parameters(), cfg.algo.world_model.optimizer)
    ao = make_optimizer(actor.parameters(), cfg.algo.actor.optimizer)
    co = make_optimizer(critic.parameters(), cfg.algo.critic.optimizer)
    step = mod._capture_train_step(runtime, wm, actor, critic, target_critic,
                                   wo, ao, co, data, cfg, False, [A])
    assert step is not None, "capture failed"
    return step


def build_dv1(runtime, cfg, A, data, mod):
    from sheeprl_amd.algos.dreamer_v1.dreamer_v1 import build_agent
    from sheeprl_amd.optim import make_optimizer

    obs_space = _space()
    wm, actor, critic, _ = build_agent(runtime, [A], False, cfg, obs_space)
    wo = make_optimizer(wm.parameters(), cfg.algo.world_model.optimizer)
    ao = make_optimizer(actor.parameters(), cfg.algo.actor.optimizer)
    co = make_optimizer(critic.parameters(), cfg.algo.critic.optimizer)
    step = mod._capture_train_step(runtime, wm, actor, critic, wo, ao, co, data, cfg)
    assert step is not None, "capture failed"
    return step


def _space():
    from sheeprl_amd.envs import spaces

    return spaces.Dict({"rgb": spaces.Box(0, 255, (3, 64, 64), np.uint8)})


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "both"
    if which in ("dv2", "both"):
        import sheeprl_amd.algos.dreamer_v2.dreamer_v2 as dv2

        rate("dreamer_v2", dv2, build_dv2)
    if which in ("dv1", "both"):
        import sheeprl_amd.algos.dreamer_v1.dreamer_v1 as dv1

        rate("dreamer_v1", dv1, build_dv1)
