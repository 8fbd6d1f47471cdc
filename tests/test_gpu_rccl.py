"""Real-RCCL multi-rank proof on one GPU.

Runs tests/helpers/rccl_2rank.py under torchrun with 2 ranks sharing cuda:0:
GradSync param broadcast, bucketed async all-reduce inside backward
(channels_last module), and a hipGraph-captured step with the collective
inside the graph.  This is the first-contact check for the driver's 8-GPU
scaling window (bench.py --gpus N uses the same GradSync + graph path).
"""

import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

REPO = Path(__file__).resolve().parents[1]


@pytest.mark.gpu
@pytest.mark.timeout(600)
def test_rccl_2rank_gradsync_graph():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env["MASTER_ADDR"] = "127.0.0.1"
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr=127.0.0.1", "--master-port=29517",
            str(REPO / "tests" / "helpers" / "rccl_2rank.py"),
        ],
        cwd=str(REPO),
        env=env,
        capture_output=True,
        text=True,
        timeout=540,
    )
    assert proc.returncode == 0, f"stdout:\n{proc.stdout}\nstderr:\n{proc.stderr}"
    assert "RCCL_2RANK_OK" in proc.stdout, proc.stdout


@pytest.mark.gpu
@pytest.mark.timeout(900)
def test_bench_2rank_torchrun_rccl(tmp_path):
    """bench.py --gpus 2 under torchrun on ONE device: the exact launch
    contract the driver uses for SCALE_rNN, on real RCCL."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env["SHEEPRL_AMD_BENCH_DEVICE_OVERSUBSCRIBE"] = "1"  # 2 ranks, 1 GPU
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr=127.0.0.1", "--master-port=29519",
            str(REPO / "bench.py"), "--gpus", "2", "--steps", "3", "--warmup", "1",
            "--override", "algo=dreamer_v3_XS",
            "--override", "algo.per_rank_batch_size=4",
            "--override", "algo.per_rank_sequence_length=8",
            "--override", "buffer.size=256",
        ],
        cwd=str(tmp_path),
        env=env,
        capture_output=True,
        text=True,
        timeout=840,
    )
    assert proc.returncode == 0, f"stdout:\n{proc.stdout}\nstderr:\n{proc.stderr}"
    import json

    lines = [l for l in proc.stdout.splitlines() if l.strip().startswith("{")]
    assert lines, proc.stdout
    rec = json.loads(lines[-1])
    assert rec["n_gpus"] == 2
