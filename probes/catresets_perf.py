import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
"""Time the fused-scan FORWARD at bench dims (DV3-S) in a hipGraph,
cat_st_resets fusion on vs off."""
import torch
from sheeprl_amd.algos.dreamer_v3.agent import RSSM, RecurrentModel
from sheeprl_amd.models import MLP
from sheeprl_amd.ops import scan as scan_mod

T, B, E, A, H, S, K, DU, P = 64, 16, 4096, 6, 512, 32, 32, 512, 1024
SK = S * K
torch.manual_seed(0)
rssm = RSSM(
    RecurrentModel(SK + A, H, DU),
    MLP(E + H, SK, [P], activation="silu", layer_norm=True, layer_norm_eps=1e-3),
    MLP(H, SK, [P], activation="silu", layer_norm=True, layer_norm_eps=1e-3),
    discrete=K, unimix=0.01,
).cuda().to(torch.bfloat16)
embed = torch.randn(T, B, E, device="cuda", dtype=torch.bfloat16)
actions = torch.randn(T, B, A, device="cuda", dtype=torch.bfloat16)
is_first = (torch.rand(T, B, 1, device="cuda") < 0.05).float()
is_first[0] = 1.0
urand = torch.rand(T, B, S, K, device="cuda")
_ir, _ip = rssm.get_initial_states((1, B))
init = (_ir.detach().contiguous(), _ip.detach().contiguous())

def fwd():
    with torch.no_grad():
        return scan_mod.rssm_scan(rssm, embed, actions, is_first, init, urand=urand)

for mode in ("1", "0"):
    os.environ["SHEEPRL_AMD_NO_CATRESETS"] = mode
    for _ in range(3):
        fwd()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        fwd()
    torch.cuda.current_stream().wait_stream(s)
    with torch.cuda.graph(g):
        fwd()
    for _ in range(5):
        g.replay()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(True); t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(50):
        g.replay()
    t1.record()
    torch.cuda.synchronize()
    label = "standalone-resets" if mode == "1" else "fused-cat_st_resets"
    print(f"{label}: {t0.elapsed_time(t1)/50*1000:.1f} us/scan-fwd")
