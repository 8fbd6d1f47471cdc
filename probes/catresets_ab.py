import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
"""A/B the cat_st_resets fusion against the standalone resets kernel: same
philox-free urand, compare x_s/hu_s assembly via outputs and grads."""
import os
import torch
from sheeprl_amd.algos.dreamer_v3.agent import RSSM, RecurrentModel
from sheeprl_amd.models import MLP
from sheeprl_amd.ops import scan as scan_mod

T, B, E, A, H, S, K, DU, P = 4, 16, 72, 6, 64, 2, 32, 64, 64
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
is_first = (torch.rand(T, B, 1, device="cuda") < 0.3).float()
is_first[0] = 1.0
urand = torch.full((T, B, S, K), 0.3, device="cuda")
pick = torch.randint(0, K, (T, B, S), device="cuda")
urand.scatter_(-1, pick.unsqueeze(-1), 1.0 - 1e-7)
gh = torch.randn(T, B, H, device="cuda", dtype=torch.bfloat16)
gz = torch.randn(T, B, SK, device="cuda", dtype=torch.bfloat16)
gm = torch.randn(T, B, SK, device="cuda")

def run(no_fuse):
    os.environ["SHEEPRL_AMD_NO_CATRESETS"] = "1" if no_fuse else "0"
    for p in rssm.parameters():
        p.grad = None
    _ir, _ip = rssm.get_initial_states((1, B))
    init = (_ir.contiguous(), _ip.contiguous())
    h, z, m = scan_mod.rssm_scan(rssm, embed, actions, is_first, init, urand=urand)
    torch.autograd.backward([h, z, m], [gh, gz, gm])
    return h.detach().clone(), z.detach().clone(), m.detach().clone(), {
        n: p.grad.clone() for n, p in rssm.named_parameters() if p.grad is not None}

h1, z1, m1, g1 = run(True)   # standalone resets (old path)
h2, z2, m2, g2 = run(False)  # fused
print("h diff", (h2 - h1).abs().max().item())
print("z equal", torch.equal(z2, z1))
print("m diff", (m2 - m1).abs().max().item())
for n in g1:
    a, b = g1[n].float(), g2[n].float()
    rel = ((a - b).norm() / a.norm().clamp_min(1e-3)).item()
    if rel > 1e-6:
        print(f"{n:45s} rel {rel:.4f}")
print("done")
