"""The driver's bench contract: torchrun-launched bench.py must emit exactly
one valid JSON line from rank 0 (CPU/gloo here; the driver runs the same
invocation on MI355X boxes)."""

import json
import os
import subprocess
import sys

import pytest

TINY = [
    "--override", "algo.world_model.recurrent_model.recurrent_state_size=64",
    "--override", "algo.world_model.transition_model.hidden_size=64",
    "--override", "algo.world_model.representation_model.hidden_size=64",
    "--override", "algo.world_model.encoder.cnn_channels_multiplier=4",
    "--override", "algo.dense_units=64",
    "--override", "algo.mlp_layers=1",
    "--override", "algo.world_model.discrete_size=8",
    "--override", "algo.world_model.stochastic_size=8",
    "--override", "algo.per_rank_batch_size=2",
    "--override", "algo.per_rank_sequence_length=8",
    "--override", "algo.horizon=3",
    "--override", "env.num_envs=1",
]


@pytest.mark.timeout(600)
def test_bench_two_rank_gloo_emits_single_json(tmp_path):
    env = dict(os.environ)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env["PYTHONPATH"] = f"{repo}:{env.get('PYTHONPATH', '')}"
    env["MASTER_ADDR"] = "127.0.0.1"
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", "29611",
         os.path.join(repo, "bench.py"), "--gpus", "2", "--steps", "2", "--warmup", "1", *TINY],
        capture_output=True, text=True, env=env, cwd=str(tmp_path), timeout=570,
    )
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    json_lines = [l for l in out.stdout.splitlines() if l.strip().startswith("{")]
    assert len(json_lines) == 1, out.stdout
    rec = json.loads(json_lines[0])
    assert rec["metric"] == "env_frames_per_sec"
    assert rec["n_gpus"] == 2 and rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["value"] > 0 and rec["higher_is_better"] is True
    assert rec["scaling"] == "weak" and rec["data"] == "synthetic"
    assert "ms_per_step" in rec and "config" in rec


@pytest.mark.timeout(600)
def test_bench_single_rank_default(tmp_path):
    """Plain `python bench.py` (the driver's N=1 run) with tiny overrides."""
    env = dict(os.environ)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env["PYTHONPATH"] = f"{repo}:{env.get('PYTHONPATH', '')}"
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"), "--steps", "2", "--warmup", "1", *TINY],
        capture_output=True, text=True, env=env, cwd=str(tmp_path), timeout=570,
    )
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    json_lines = [l for l in out.stdout.splitlines() if l.strip().startswith("{")]
    assert len(json_lines) == 1, out.stdout
    rec = json.loads(json_lines[0])
    assert rec["n_gpus"] == 1 and rec["metric"] == "env_frames_per_sec"


@pytest.mark.timeout(600)
def test_bench_ppo_algo_single_rank(tmp_path):
    """`python bench.py --algo ppo` emits one valid contract JSON line."""
    env = dict(os.environ)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env["PYTHONPATH"] = f"{repo}:{env.get('PYTHONPATH', '')}"
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"), "--algo", "ppo", "--steps", "1",
         "--warmup", "0", "--override", "algo.rollout_steps=8", "--override", "env.num_envs=2",
         "--override", "algo.per_rank_batch_size=8", "--override", "algo.update_epochs=1"],
        capture_output=True, text=True, env=env, cwd=str(tmp_path), timeout=570,
    )
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    json_lines = [l for l in out.stdout.splitlines() if l.strip().startswith("{")]
    assert len(json_lines) == 1, out.stdout
    rec = json.loads(json_lines[0])
    assert rec["metric"] == "env_frames_per_sec" and rec["config"]["model"] == "ppo_pixel"
    assert rec["value"] > 0 and rec["scaling"] == "weak"
