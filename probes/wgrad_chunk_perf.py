import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
"""Time the fused scan fwd+bwd in a hipGraph with interleaved weight-grad
chunk sizes (SHEEPRL_AMD_WGRAD_CHUNK = 0 -> single whole-sequence GEMMs)."""
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
embed = torch.randn(T, B, E, device="cuda", dtype=torch.bfloat16, requires_grad=True)
actions = torch.randn(T, B, A, device="cuda", dtype=torch.bfloat16)
is_first = (torch.rand(T, B, 1, device="cuda") < 0.05).float()
is_first[0] = 1.0
urand = torch.rand(T, B, S, K, device="cuda")
gh = torch.randn(T, B, H, device="cuda", dtype=torch.bfloat16)
gz = torch.randn(T, B, SK, device="cuda", dtype=torch.bfloat16)
gm = torch.randn(T, B, SK, device="cuda")
_ir, _ip = rssm.get_initial_states((1, B))
init = (_ir.detach().contiguous(), _ip.detach().contiguous())

def step():
    for p in rssm.parameters():
        p.grad = None
    h, z, m = scan_mod.rssm_scan(rssm, embed, actions, is_first, init, urand=urand)
    torch.autograd.backward([h, z, m], [gh, gz, gm])

for chunk in ("0", "32", "16", "8"):
    os.environ["SHEEPRL_AMD_WGRAD_CHUNK"] = chunk
    for _ in range(3):
        step()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        step()
    torch.cuda.current_stream().wait_stream(s)
    with torch.cuda.graph(g):
        step()
    for _ in range(5):
        g.replay()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(True); t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(30):
        g.replay()
    t1.record()
    torch.cuda.synchronize()
    print(f"wgrad_chunk={chunk}: {t0.elapsed_time(t1)/30:.3f} ms/scan fwd+bwd")
