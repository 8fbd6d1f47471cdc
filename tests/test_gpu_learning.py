"""Learning-quality composition guard: the graphed-bf16
fused path must track the eager-fp32 module path's world-model loss curve on
identical synthetic data (shared eager-fp32 evaluator on a held-out batch)."""

import json
import subprocess
import sys
from pathlib import Path

import pytest
import torch

REPO = Path(__file__).resolve().parents[1]


@pytest.mark.gpu
@pytest.mark.timeout(900)
def test_loss_curve_parity_fused_vs_eager(tmp_path):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    out = tmp_path / "loss_parity.json"
    proc = subprocess.run(
        [sys.executable, str(REPO / "probes" / "loss_parity.py"), "300", str(out)],
        cwd=str(REPO),
        capture_output=True,
        text=True,
        timeout=840,
    )
    assert proc.returncode == 0, f"stdout:\n{proc.stdout[-3000:]}\nstderr:\n{proc.stderr[-3000:]}"
    res = json.loads(out.read_text())
    assert res["both_learning"], res
    assert res["rel_final_divergence"] < 0.15, res
