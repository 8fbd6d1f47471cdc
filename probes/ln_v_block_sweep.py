"""Microbench: ln_act fwd/bwd on the behaviour-MLP shape [16384, 1024] bf16.

The wave-per-row vectorized backward measures ~98 us in the XL graph
(~1 TB/s effective) — ~6x off the traffic roofline.  This sweeps the
batch shape against torch-native composites to separate kernel-internal
limits from trace effects, and times the op at several row counts.

Run on a GPU box:  python probes/ln_v_block_sweep.py
"""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from sheeprl_amd import ops


def bench(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    assert torch.cuda.is_available()
    for N, D in [(16384, 1024), (4096, 1024), (16384, 768), (65536, 1024)]:
        x = torch.randn(N, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        w = torch.rand(D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        b = torch.randn(D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        g = torch.randn(N, D, device="cuda", dtype=torch.bfloat16)

        y = ops.layer_norm_act(x, w, b, 1e-3, "silu")
        t_fwd = bench(lambda: ops.layer_norm_act(x.detach(), w.detach(), b.detach(), 1e-3, "silu"))

        def bwd():
            xg = x.detach().requires_grad_()
            out = ops.layer_norm_act(xg, w, b, 1e-3, "silu")
            out.backward(g)

        t_full = bench(bwd, iters=100)

        # torch-native reference (LN fp32 + SiLU) for a bandwidth yardstick
        ln = torch.nn.LayerNorm(D, eps=1e-3).cuda()
        xf = x.detach().float()
        t_torch_fwd = bench(lambda: torch.nn.functional.silu(ln(xf)))

        mb = N * D * 2 / 1e6
        print(
            f"[{N}x{D}] fwd {t_fwd:7.1f} us ({3*mb/t_fwd*1e3/1e3:.1f} TB/s eff)  "
            f"fwd+bwd {t_full:7.1f} us   torch-fp32 LN+SiLU fwd {t_torch_fwd:7.1f} us"
        )


if __name__ == "__main__":
    main()
