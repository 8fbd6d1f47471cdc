"""Stable-Baselines3 comparison harness (parity: benchmarks/benchmark_sb3.py
in the reference, which times SB3's PPO/SAC on the same workloads as the
`*_benchmarks` experiment presets).

Run the two sides on the same machine and compare wall time:

    python benchmarks/benchmark_sb3.py ppo      # SB3 side (needs sb3 + gymnasium)
    python benchmarks/benchmark.py exp=ppo_benchmarks   # our side

This offline image ships neither stable_baselines3 nor gymnasium, so the SB3
side degrades to an actionable message; the harness activates unchanged once
the packages are installed.  Workload shapes mirror BASELINE.md's SB3 column:
PPO CartPole-v1 65 536 steps (rollout 128, batch 64) and SAC
LunarLanderContinuous-v2 65 536 steps (4 envs, batch 256).
"""

import sys
import time

WORKLOADS = {
    "ppo": dict(env_id="CartPole-v1", total=65_536,
                kwargs=dict(n_steps=128, batch_size=64)),
    "sac": dict(env_id="LunarLanderContinuous-v2", total=65_536,
                kwargs=dict(batch_size=256)),
}


def main() -> int:
    algo = (sys.argv[1] if len(sys.argv) > 1 else "ppo").lower()
    if algo not in WORKLOADS:
        print(f"unknown workload {algo!r}; choose from {sorted(WORKLOADS)}")
        return 2
    try:
        import gymnasium
        import stable_baselines3 as sb3
    except ImportError as e:
        print(
            f"SB3 side unavailable in this image ({e}).\n"
            "Install stable-baselines3 + gymnasium to produce the comparison "
            "column; the sheeprl-amd side runs offline via\n"
            f"    python benchmarks/benchmark.py exp={algo}_benchmarks"
        )
        return 1
    w = WORKLOADS[algo]
    env = gymnasium.make(w["env_id"])
    model = {"ppo": sb3.PPO, "sac": sb3.SAC}[algo]("MlpPolicy", env, **w["kwargs"])
    tic = time.perf_counter()
    model.learn(total_timesteps=w["total"], progress_bar=False)
    print(f"SB3 {algo} {w['env_id']} {w['total']} steps: "
          f"{time.perf_counter() - tic:.2f} s wall")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
