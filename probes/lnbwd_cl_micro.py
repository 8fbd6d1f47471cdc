import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
"""Micro-time ln_act_bwd on the bench's channels-last/MLP LN shapes, in-graph."""
import torch
from sheeprl_amd.ops._ext import require_ext

ext = require_ext()
shapes = [(1024, 512), (16384, 512), (1024, 1024), (65536, 128), (1048576, 32)]
torch.manual_seed(0)
tensors = []
for R, D in shapes:
    x = torch.randn(R, D, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    y, mean, rstd = ext.ln_act_fwd(x, w, b, 1e-3, True)
    gy = torch.randn_like(x)
    gw = torch.zeros(D, device="cuda", dtype=torch.float32)
    gb = torch.zeros(D, device="cuda", dtype=torch.float32)
    tensors.append((x, w, b, mean, rstd, gy, gw, gb))

def run_all():
    for (x, w, b, mean, rstd, gy, gw, gb) in tensors:
        ext.ln_act_bwd(gy, x, w, b, mean, rstd, True)

run_all(); torch.cuda.synchronize()
g = torch.cuda.CUDAGraph()
s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    run_all()
torch.cuda.current_stream().wait_stream(s)
with torch.cuda.graph(g):
    run_all()
for _ in range(5):
    g.replay()
torch.cuda.synchronize()
t0 = torch.cuda.Event(True); t1 = torch.cuda.Event(True)
t0.record()
for _ in range(100):
    g.replay()
t1.record(); torch.cuda.synchronize()
print(f"floor={os.environ.get('SHEEPRL_AMD_LN_BWD_FLOOR','64')}: "
      f"{t0.elapsed_time(t1)/100*1000:.1f} us for {len(shapes)} bwd calls")
