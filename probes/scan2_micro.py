"""Micro-benchmark of the scan2 fused phase kernels vs the round-1 launch
pairs they replace (S-model shapes).  Run on the GPU box:

    python probes/scan2_micro.py
"""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from sheeprl_amd.ops._ext import require_ext

ext = require_ext()
dev = "cuda"
dt = torch.bfloat16

B, T = 16, 64
H, D, P, SK, A, E, KD = 512, 512, 512, 1024, 6, 4096, 32
eps = 1e-3
unimix = 0.01

g = torch.Generator(device=dev).manual_seed(0)


def bf(*shape):
    return torch.randn(*shape, generator=g, device=dev, dtype=dt)


def timeit(name, fn, reps=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record()
    torch.cuda.synchronize()
    print(f"{name:28s} {t0.elapsed_time(t1) / reps * 1000:8.2f} us")


# ---- inputs
z_prev, iz = bf(B, SK), bf(B, SK)
h_prev, ih = bf(B, H), bf(B, H)
act = bf(B, A)
f = (torch.rand(B, generator=g, device=dev) < 0.2).to(dt)
w1 = bf(D, SK + A)
lnw1, lnb1 = bf(D), bf(D)
w2 = bf(3 * H, H + D)
lnwg, lnbg = bf(3 * H), bf(3 * H)
w3 = bf(P, H + E)
lnw3, lnb3 = bf(P), bf(P)
w4 = bf(SK, P)
b4 = bf(SK)
x_s = torch.empty(B, SK + A, device=dev, dtype=dt)
hu_s = torch.empty(B, H + D, device=dev, dtype=dt)
g1_s = torch.empty(B, D, device=dev, dtype=dt)
y_s = torch.empty(B, 3 * H, device=dev, dtype=dt)
r_s = torch.empty(B, H + E, device=dev, dtype=dt)
g3_s = torch.empty(B, P, device=dev, dtype=dt)
p_s = torch.empty(B, P, device=dev, dtype=dt)
h_seq = torch.empty(B, H, device=dev, dtype=dt)
z_seq = torch.empty(B, SK, device=dev, dtype=dt)
m_seq = torch.empty(B, SK, device=dev, dtype=torch.float32)
s_s = torch.empty(B, SK, device=dev, dtype=torch.float32)
mr = torch.empty(8, B, device=dev, dtype=torch.float32)
urand = torch.rand(B, SK, generator=g, device=dev)
r_s[:, H:] = bf(B, E)

ws2 = torch.zeros(64, 32, device=dev, dtype=torch.float32)
tk = torch.zeros(64, device=dev, dtype=torch.int32)
slot = [0]


def next_slot():
    # amortize the slot zeroing: one bulk zero per 64 calls (2/64 launches
    # per call instead of 2 — the real scan zeroes all T slots once per fwd)
    slot[0] = (slot[0] + 1) % 64
    if slot[0] == 0:
        ws2.zero_()
        tk.zero_()
    return slot[0]


# fwd fused
def f1():
    s = next_slot()
    ext.scan2_f1(z_prev, iz, h_prev, ih, act, f, w1, lnw1, lnb1,
                 x_s, hu_s, g1_s, mr[0], mr[1], ws2[s], tk[s], eps)


def f2():
    s = next_slot()
    ext.scan2_f2(hu_s, w2, lnwg, lnbg, y_s, h_seq, r_s[:, :H], mr[2], mr[3], ws2[s], tk[s], eps)


def f3():
    s = next_slot()
    ext.scan2_f3(r_s, w3, lnw3, lnb3, p_s, g3_s, mr[4], mr[5], ws2[s], tk[s], eps)


def f4():
    ext.scan2_f4(p_s, w4, b4, urand, m_seq, z_seq, s_s, KD, unimix)


# round-1 equivalents
w1t, w2t, w3t, w4t = w1.t(), w2.t(), w3.t(), w4.t()
g1b = torch.empty(B, D, device=dev, dtype=dt)
yb = torch.empty(B, 3 * H, device=dev, dtype=dt)
g3b = torch.empty(B, P, device=dev, dtype=dt)
rawb = torch.empty(B, SK, device=dev, dtype=dt)
x_c = x_s.clone()
hu_c = hu_s.clone()
sk_scratch = torch.zeros(16, P, device=dev, dtype=torch.float32)
sk_tickets = torch.zeros(P // 64, device=dev, dtype=torch.int32)


def v1_f1():
    ext.scan_resets_fwd(z_prev, iz, h_prev, ih, act, f, x_c, hu_c, False)
    torch.mm(x_c, w1t, out=g1b)
    ext.ln_act_fwd_o(g1b, lnw1, lnb1, eps, True, hu_c[:, H:], mr[0], mr[1])


def v1_f2():
    torch.mm(hu_c, w2t, out=yb)
    ext.gru_gates_fwd_o(yb, hu_c[:, :H], lnwg, lnbg, eps, h_seq, r_s[:, :H], mr[2], mr[3])


def v1_f3():
    ext.g16_splitk(r_s, w3, None, sk_scratch, sk_tickets, g3b, 5)
    ext.ln_act_fwd_o(g3b, lnw3, lnb3, eps, True, p_s, mr[4], mr[5])


def v1_f3_blas():
    torch.mm(r_s, w3t, out=g3b)
    ext.ln_act_fwd_o(g3b, lnw3, lnb3, eps, True, p_s, mr[4], mr[5])


def v1_f4():
    torch.addmm(b4, p_s, w4t, out=rawb)
    ext.cat_st_fwd_o(rawb.view(B, SK // KD, KD), urand.view(B, SK // KD, KD), unimix,
                     m_seq.view(B, SK // KD, KD), z_seq.view(B, SK // KD, KD),
                     s_s.view(B, SK // KD, KD))


# bwd fused
gm_in = torch.randn(B, SK, generator=g, device=dev)
gz_in = bf(B, SK)
gh_in = bf(B, H)
gz_c = torch.empty(B, SK, device=dev, dtype=dt)
gh_c = torch.empty(B, H, device=dev, dtype=dt)
graw_o = torch.empty(B, SK, device=dev, dtype=dt)
gp_o = torch.empty(B, P, device=dev, dtype=dt)
gg3_o = torch.empty(B, P, device=dev, dtype=dt)
gr_o = torch.empty(B, H + E, device=dev, dtype=dt)
gy_o = torch.empty(B, 3 * H, device=dev, dtype=dt)
ghu_o = torch.empty(B, H + D, device=dev, dtype=dt)
gg1_o = torch.empty(B, D, device=dev, dtype=dt)
ga_o = torch.empty(B, A, device=dev, dtype=dt)
accs = torch.zeros(2 * D + 6 * H + 2 * P + B * (H + SK), device=dev, dtype=torch.float32)
glnw1a, glnb1a = accs[:D], accs[D:2 * D]
glnwga, glnbga = accs[2 * D:2 * D + 3 * H], accs[2 * D + 3 * H:2 * D + 6 * H]
glnw3a, glnb3a = accs[2 * D + 6 * H:2 * D + 6 * H + P], accs[2 * D + 6 * H + P:2 * D + 6 * H + 2 * P]
gih_a = accs[2 * D + 6 * H + 2 * P:2 * D + 6 * H + 2 * P + B * H].view(B, H)
giz_a = accs[-B * SK:].view(B, SK)
w1tc, w2tc, w3tc, w4tc = [w.t().contiguous() for w in (w1, w2, w3, w4)]
s_s.uniform_(0.01, 1.0)


def pb4():
    ext.scan2_b4(gm_in, gz_in, gz_c, s_s, w4tc, graw_o, gp_o, KD, unimix)


def pb3():
    ext.scan2_b3(gp_o, g3_s, lnw3, lnb3, mr[4], mr[5], w3tc, gg3_o, glnw3a, glnb3a, gr_o)


def pb2():
    ext.scan2_b2(gh_in, gh_c, gr_o[:, :H], y_s, hu_s[:, :H], lnwg, lnbg, mr[2], mr[3],
                 w2tc, f, gy_o, glnwga, glnbga, gh_c, gih_a, ghu_o)


def pb1():
    ext.scan2_b1(ghu_o[:, H:], g1_s, lnw1, lnb1, mr[0], mr[1], w1tc, f, gg1_o,
                 glnw1a, glnb1a, gz_c, giz_a, ga_o)


def v1_b4():
    ext.cat_st_bwd_o(gm_in.view(B, -1, KD), gz_in.view(B, -1, KD), gz_c.view(B, -1, KD),
                     s_s.view(B, -1, KD), unimix, graw_o.view(B, -1, KD))
    torch.mm(graw_o, w4, out=gp_o)


def v1_b3():
    ext.ln_act_bwd_acc(gp_o, g3_s, lnw3, lnb3, mr[4], mr[5], True, gg3_o, glnw3a, glnb3a)
    torch.mm(gg3_o, w3, out=gr_o)


def v1_b2():
    ext.gru_gates_bwd_acc(gh_in, gh_c, gr_o[:, :H], y_s, hu_s[:, :H], lnwg, lnbg,
                          mr[2], mr[3], gy_o, gh_c, glnwga, glnbga)
    torch.mm(gy_o, w2, out=ghu_o)


def v1_b1():
    ext.ln_act_bwd_acc(ghu_o[:, H:], g1_s, lnw1, lnb1, mr[0], mr[1], True, gg1_o, glnw1a, glnb1a)
    torch.mm(gg1_o, w1, out=gg1_o.new_empty(B, SK + A))
    ext.scan_resets_bwd(ghu_o, gh_c, x_s, f, gh_c, gz_c, ga_o, gih_a, giz_a)


print("== forward phases (fused vs round-1 launch pairs) ==")
timeit("scan2_f1", f1)
timeit("v1 f1 (resets+mm+ln)", v1_f1)
timeit("scan2_f2", f2)
timeit("v1 f2 (mm+gru)", v1_f2)
timeit("scan2_f3", f3)
timeit("v1 f3 (splitk+ln)", v1_f3)
timeit("v1 f3 (hipblaslt+ln)", v1_f3_blas)
timeit("scan2_f4", f4)
timeit("v1 f4 (addmm+catst)", v1_f4)
print("== backward phases ==")
timeit("scan2_b4", pb4)
timeit("v1 b4 (catst_bwd+mm)", v1_b4)
timeit("scan2_b3", pb3)
timeit("v1 b3 (lnbwd+mm)", v1_b3)
timeit("scan2_b2", pb2)
timeit("v1 b2 (grubwd+mm)", v1_b2)
timeit("scan2_b1", pb1)
timeit("v1 b1 (lnbwd+mm+resets)", v1_b1)

# ---- scan3 split-K variants (generation tickets: monotonic per call)
scr1 = torch.zeros(16, D, device=dev, dtype=torch.float32)
scr2 = torch.zeros(16, 3 * H, device=dev, dtype=torch.float32)
scr3 = torch.zeros(16, P, device=dev, dtype=torch.float32)
scr4 = torch.zeros(16, SK, device=dev, dtype=torch.float32)
scrb4 = torch.zeros(16, P, device=dev, dtype=torch.float32)
tk3 = torch.zeros(4, 256, device=dev, dtype=torch.int32)
tk2g = torch.zeros(4, device=dev, dtype=torch.int32)
tkb4 = torch.zeros(P // 16, device=dev, dtype=torch.int32)
ws3 = torch.zeros(4, 32, device=dev, dtype=torch.float32)
gen_c = [0, 0, 0, 0, 0]


def s3f1():
    gen_c[0] += 1
    ws3[0].zero_()
    ext.scan3_f1(z_prev, iz, h_prev, ih, act, f, w1, lnw1, lnb1, x_s, hu_s, g1_s,
                 mr[0], mr[1], scr1, tk3[0], tk2g[0:1], ws3[0], eps, gen_c[0])


def s3f2():
    gen_c[1] += 1
    ws3[1].zero_()
    ext.scan3_f2(hu_s, w2, lnwg, lnbg, y_s, h_seq, r_s[:, :H], mr[2], mr[3],
                 scr2, tk3[1], tk2g[1:2], ws3[1], eps, gen_c[1])


def s3f3():
    gen_c[2] += 1
    ws3[2].zero_()
    ext.scan3_f3(r_s, w3, lnw3, lnb3, p_s, g3_s, mr[4], mr[5], scr3, tk3[2],
                 tk2g[2:3], ws3[2], eps, gen_c[2])


def s3f4():
    gen_c[3] += 1
    ext.scan3_f4(p_s, w4, b4, urand, m_seq, z_seq, s_s, scr4, tk3[3], KD, unimix, gen_c[3])


def s3b4():
    gen_c[4] += 1
    ext.scan3_b4(gm_in, gz_in, gz_c, s_s, w4tc, graw_o, gp_o, scrb4, tkb4, KD, unimix, gen_c[4])


print("== scan3 split-K variants ==")
timeit("scan3_f1", s3f1)
timeit("scan3_f2", s3f2)
timeit("scan3_f3", s3f3)
timeit("scan3_f4", s3f4)
timeit("scan3_b4", s3b4)

# ---- weight-grad GEMM variants (the 5 batched [out, TB] x [TB, in] GEMMs
# measured ~100 us each in-graph as mm(g.t(), x))
TB = 1024
gg3_big = bf(TB, P)
r_big = bf(TB, H + E)
gy_big = bf(TB, 3 * H)
hu_big = bf(TB, H + D)


def wg_tn():
    torch.mm(gg3_big.t(), r_big)
    torch.mm(gy_big.t(), hu_big)


def wg_tr_nn():
    torch.mm(ext.transpose2d(gg3_big), r_big)
    torch.mm(ext.transpose2d(gy_big), hu_big)


def wg_nt_t():
    torch.mm(r_big.t(), gg3_big).t()
    torch.mm(hu_big.t(), gy_big).t()


print("== weight-grad GEMM variants (2 GEMMs each) ==")
timeit("mm(g.t(), x)  TN", wg_tn)
timeit("transpose2d + NN", wg_tr_nn)
timeit("mm(x.t(), g).t()", wg_nt_t)
