"""End-to-end learning quality on CPU: PPO must actually
solve CartPole through the real CLI — the reference's headline rows pair
wall-clock with reward, so throughput tests alone don't guard the math.

~50 s on 4 CPU cores (65 536 steps, the reference benchmark budget,
BASELINE.md "PPO 65 536 steps").
"""

import json
import os
from pathlib import Path

import pytest


@pytest.mark.timeout(900)
def test_ppo_cartpole_solves(tmp_path):
    from sheeprl_amd.cli import run

    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        run([
            "exp=ppo",
            "algo.total_steps=65536",
            "seed=7",
            "metric.log_every=16384",
            "algo.anneal_lr=True",
            "algo.max_grad_norm=0.5",
            "checkpoint.every=0",
            "checkpoint.save_last=False",
            "algo.run_test=True",
        ])
    finally:
        os.chdir(cwd)
    metrics = sorted(Path(tmp_path).glob("logs/runs/**/metrics.jsonl"))
    assert metrics, "no metrics written"
    rows = [json.loads(l) for l in metrics[-1].read_text().splitlines()]
    test_rewards = [r["Test/cumulative_reward"] for r in rows if "Test/cumulative_reward" in r]
    assert test_rewards, rows[-3:]
    # reference parity bar: CartPole solved (max return 500)
    assert test_rewards[-1] >= 475.0, f"greedy test reward {test_rewards[-1]}"
