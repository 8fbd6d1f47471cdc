"""Fused RSSM sequence scan (forward + hand-written backward).

The DV3 dynamic-learning scan is T=64 sequential steps; under autograd each
step leaves ~40 bookkeeping kernels in the backward (per-use gradient
accumulation adds on every weight, cat-backward copies, zero fills).  This
Function runs the whole scan with explicit kernels and a launch-lean layout:

* every per-step intermediate is written DIRECTLY into its slice of a stacked
  [T, B, *] buffer (strided-out kernel variants, ``mm(..., out=)``) — no
  per-step stacking copies, no ``torch.cat`` (the concatenated GEMM inputs
  are assembled in place: resets write into ``x_s``/``hu_s`` column blocks,
  the GRU writes ``h`` into both ``h_seq`` and the representation input
  ``r_s`` whose embed columns are filled once, batched, before the loop);
* the backward reverse loop stores per-step output gradients into stacked
  buffers and accumulates LN affine / init-state gradients into fp32 buffers
  INSIDE the kernels (``*_acc`` variants) — zero per-step fills/casts/adds;
* all weight gradients are then computed AFTER the loop as four big
  [out, T*B] x [T*B, in] GEMMs (MFMA-sized) instead of 4x64 latency-bound
  rank-B ``addmm_`` updates.

Per step (math identical to RSSM.dynamic_posterior + RecurrentModel +
LayerNormGRUCell + representation MLP + the fused categorical-ST head):

    a' , h', z'  = masked_lerp(action/h/z, init, is_first)
    u  = silu(LN(W1 [z', a']))                 (recurrent_model.mlp)
    y  = W2 [h', u]                            (GRU input projection)
    h  = gru_gates(y, h')                      (LN + Hafner gates)
    p  = silu(LN(W3 [h, embed]))               (representation hidden)
    raw= W4 p + b4
    m, z = categorical_st(raw)                 (unimix log-probs + ST sample)

Used on CUDA only (the CPU path keeps the module-based loop); validated
against the module loop on GPU in tests/test_gpu_kernels.py.
"""

from __future__ import annotations

from typing import Any, Tuple

import os

import torch
from torch import Tensor

from sheeprl_amd.ops._ext import require_ext


def _scan_impl() -> str:
    """Fused-phase scan implementation: "v1" (launch-per-op with hipblaslt
    GEMMs — fastest measured in-graph; the guide's "many small graphed
    launches win" rule), "v2" (N-split + ticket LN) or "v3" (split-K +
    generation tickets), kept as measured experiments."""
    return os.environ.get("SHEEPRL_AMD_SCAN_IMPL", "v1")


class _RSSMScan(torch.autograd.Function):
    @staticmethod
    def forward(
        ctx,
        embed: Tensor,       # [T, B, E]
        actions: Tensor,     # [T, B, A]
        is_first: Tensor,    # [T, B, 1]
        init_h: Tensor,      # [1, B, H]
        init_z: Tensor,      # [1, B, SK]
        w1: Tensor,          # [D, SK+A]
        lnw1: Tensor, lnb1: Tensor,
        w2: Tensor,          # [3H, H+D]
        lnwg: Tensor, lnbg: Tensor,
        w3: Tensor,          # [P, H+E]
        lnw3: Tensor, lnb3: Tensor,
        w4: Tensor,          # [SK, P]
        b4: Tensor,          # [SK]
        urand_all,           # [T, B, S, K] fp32 or None (test parity hook)
        unimix: float,
        eps: float,
        discrete: int,
    ):
        ext = require_ext()
        T, B, E = embed.shape
        A = actions.shape[-1]
        H = init_h.shape[-1]
        SK = init_z.shape[-1]
        D = w1.shape[0]
        P = w3.shape[0]
        S = SK // discrete
        dev = embed.device
        dt = embed.dtype

        h_seq = torch.empty(T, B, H, device=dev, dtype=dt)
        z_seq = torch.empty(T, B, SK, device=dev, dtype=dt)
        m_seq = torch.empty(T, B, SK, device=dev, dtype=torch.float32)
        # stacked intermediates, written in place by the step kernels
        x_s = torch.empty(T, B, SK + A, device=dev, dtype=dt)     # [z', a']
        g1_s = torch.empty(T, B, D, device=dev, dtype=dt)
        mr1_s = torch.empty(2, T, B, device=dev, dtype=torch.float32)
        hu_s = torch.empty(T, B, H + D, device=dev, dtype=dt)     # [h', u]
        y_s = torch.empty(T, B, 3 * H, device=dev, dtype=dt)
        mrg_s = torch.empty(2, T, B, device=dev, dtype=torch.float32)
        r_s = torch.empty(T, B, H + E, device=dev, dtype=dt)      # [h, embed]
        mr3_s = torch.empty(2, T, B, device=dev, dtype=torch.float32)
        g3_s = torch.empty(T, B, P, device=dev, dtype=dt)
        p_s = torch.empty(T, B, P, device=dev, dtype=dt)
        s_s = torch.empty(T, B, S, discrete, device=dev, dtype=torch.float32)
        raw = torch.empty(B, SK, device=dev, dtype=dt)

        r_s[:, :, H:] = embed                 # batched, once
        f_all = is_first.to(dt).reshape(T, B)
        if urand_all is None:
            # one batched philox launch; tests pass explicit per-step-stacked
            # urand to match the module loop's rand stream exactly
            urand_all = torch.rand(T, B, S, discrete, device=dev, dtype=torch.float32)
        ih = init_h[0]
        iz = init_z[0]

        # persistent-kernel path: the whole T-step recurrence in ONE launch
        # (5 device-wide barriers per step instead of ~10 kernel launches);
        # writes the exact same stacked buffers, so backward is shared.
        pk_ok = (
            dt == torch.bfloat16
            and B <= 16
            and w1.dtype == dt
            and D % 64 == 0 and H % 64 == 0 and P % 64 == 0 and SK % 64 == 0
            and (H + D) % 32 == 0 and (H + E) % 32 == 0
            and discrete <= 64 and 64 % discrete == 0
            and os.environ.get("SHEEPRL_AMD_PK", "0") == "1"
        )
        if pk_ok:
            ws = torch.zeros(192, device=dev, dtype=torch.float32)
            ibar = torch.zeros(2, device=dev, dtype=torch.int32)
            ext.pk_scan_fwd(
                actions.contiguous(), f_all.contiguous(), ih.contiguous(), iz.contiguous(),
                w1.contiguous(), lnw1.contiguous(), lnb1.contiguous(),
                w2.contiguous(), lnwg.contiguous(), lnbg.contiguous(),
                w3.contiguous(), lnw3.contiguous(), lnb3.contiguous(),
                w4.contiguous(), b4.contiguous(), urand_all,
                x_s, g1_s, hu_s, y_s, r_s, g3_s, p_s, h_seq, z_seq,
                mr1_s, mrg_s, mr3_s, m_seq, s_s, ws, ibar, unimix, eps,
            )
            ctx.save_for_backward(
                f_all, w1, lnw1, lnb1, w2, lnwg, lnbg, w3, lnw3, lnb3, w4,
                x_s, g1_s, mr1_s, hu_s, y_s, mrg_s, r_s, g3_s, mr3_s, p_s, s_s,
            )
            ctx.dims = (T, B, E, A, H, SK, D, P, S, discrete)
            ctx.unimix = unimix
            return h_seq, z_seq, m_seq
        # round-2 fused-phase path: ONE multi-workgroup GEMM+epilogue kernel
        # per phase (4 launches/step instead of ~9; LN stats exchanged with
        # the in-launch ticket pattern).  Falls back to the launch-per-op path
        # for shapes outside the kernel contracts.
        v2_ok = (
            dt == torch.bfloat16
            and B <= 16
            and w1.dtype == dt
            and D % 64 == 0 and H % 64 == 0 and P % 64 == 0 and SK % 64 == 0
            and discrete <= 64 and (discrete & (discrete - 1)) == 0
            and (H + E) % 8 == 0
            and max(D, 3 * H, P, SK) // 64 <= 256
            and 16 * (((H + E) + 31) & ~31) * 2 + 160 <= 160 * 1024
            and 16 * 3 * H * 2 + 16 * H * 2 + 160 <= 160 * 1024
            and _scan_impl() in ("v2", "v3")
        )
        if v2_ok:
            iz_c = iz.contiguous()
            ih_c = ih.contiguous()
            if _scan_impl() == "v3":
                # split-K fused phases: hipblaslt-grade workgroup parallelism
                # with generation tickets (no per-step state resets)
                ws_f = torch.zeros(T, 3, 32, device=dev, dtype=torch.float32)
                tk3 = torch.zeros(3, 256, device=dev, dtype=torch.int32)
                tk2 = torch.zeros(3, device=dev, dtype=torch.int32)
                tk4 = torch.zeros(SK // 64, device=dev, dtype=torch.int32)
                scr1 = torch.zeros(16, D, device=dev, dtype=torch.float32)
                scr2 = torch.zeros(16, 3 * H, device=dev, dtype=torch.float32)
                scr3 = torch.zeros(16, P, device=dev, dtype=torch.float32)
                scr4 = torch.zeros(16, SK, device=dev, dtype=torch.float32)
                for t in range(T):
                    gen = t + 1
                    zp = None if t == 0 else z_seq[t - 1]
                    hp = None if t == 0 else h_seq[t - 1]
                    ext.scan3_f1(zp, iz_c, hp, ih_c, actions[t], f_all[t], w1, lnw1, lnb1,
                                 x_s[t], hu_s[t], g1_s[t], mr1_s[0, t], mr1_s[1, t],
                                 scr1, tk3[0], tk2[0:1], ws_f[t, 0], eps, gen)
                    ext.scan3_f2(hu_s[t], w2, lnwg, lnbg, y_s[t], h_seq[t], r_s[t, :, :H],
                                 mrg_s[0, t], mrg_s[1, t], scr2, tk3[1], tk2[1:2],
                                 ws_f[t, 1], eps, gen)
                    ext.scan3_f3(r_s[t], w3, lnw3, lnb3, p_s[t], g3_s[t], mr3_s[0, t],
                                 mr3_s[1, t], scr3, tk3[2], tk2[2:3], ws_f[t, 2], eps, gen)
                    ext.scan3_f4(p_s[t], w4, b4, urand_all[t].reshape(B, SK), m_seq[t],
                                 z_seq[t], s_s[t].view(B, SK), scr4, tk4, discrete, unimix, gen)
            else:
                ws_f = torch.zeros(T, 3, 32, device=dev, dtype=torch.float32)
                tk_f = torch.zeros(T, 3, device=dev, dtype=torch.int32)
                for t in range(T):
                    zp = None if t == 0 else z_seq[t - 1]
                    hp = None if t == 0 else h_seq[t - 1]
                    ext.scan2_f1(zp, iz_c, hp, ih_c, actions[t], f_all[t], w1, lnw1, lnb1,
                                 x_s[t], hu_s[t], g1_s[t], mr1_s[0, t], mr1_s[1, t],
                                 ws_f[t, 0], tk_f[t, 0], eps)
                    ext.scan2_f2(hu_s[t], w2, lnwg, lnbg, y_s[t], h_seq[t], r_s[t, :, :H],
                                 mrg_s[0, t], mrg_s[1, t], ws_f[t, 1], tk_f[t, 1], eps)
                    ext.scan2_f3(r_s[t], w3, lnw3, lnb3, p_s[t], g3_s[t], mr3_s[0, t],
                                 mr3_s[1, t], ws_f[t, 2], tk_f[t, 2], eps)
                    ext.scan2_f4(p_s[t], w4, b4, urand_all[t].reshape(B, SK), m_seq[t],
                                 z_seq[t], s_s[t].view(B, SK), discrete, unimix)
            ctx.save_for_backward(
                f_all, w1, lnw1, lnb1, w2, lnwg, lnbg, w3, lnw3, lnb3, w4,
                x_s, g1_s, mr1_s, hu_s, y_s, mrg_s, r_s, g3_s, mr3_s, p_s, s_s,
            )
            ctx.dims = (T, B, E, A, H, SK, D, P, S, discrete)
            ctx.unimix = unimix
            ctx.v2 = True
            ctx.impl = _scan_impl()
            return h_seq, z_seq, m_seq
        h0 = torch.zeros(B, H, device=dev, dtype=dt)
        z0 = torch.zeros(B, SK, device=dev, dtype=dt)
        # hand-written split-K M=16 MFMA kernel for the long-K
        # representation GEMM (the one shape where it beats hipblaslt)
        g16 = (
            dt == torch.bfloat16
            and B <= 16
            and w1.dtype == dt
            and (3 * H) % 64 == 0
            and P % 64 == 0
            and SK % 64 == 0
        )
        g16 = g16 and (H + E) >= 4096  # split-K pays off only at long K
        # long-K representation phase: the scan3 split-K GEMM+LN kernel
        # measured 12.5 us in-graph vs 16.5 for splitk+ln_act (two launches)
        g16_v3 = g16 and P % 16 == 0 and (H + E) % 8 == 0 and r_s.stride(2) == 1
        if g16_v3:
            s3_scr = torch.zeros(16, P, device=dev, dtype=torch.float32)
            s3_tk = torch.zeros(max(P // 16, 1), device=dev, dtype=torch.int32)
            s3_tk2 = torch.zeros(1, device=dev, dtype=torch.int32)
            s3_ws = torch.zeros(T, 32, device=dev, dtype=torch.float32)
        elif g16:
            sk_scratch = torch.zeros(16, P, device=dev, dtype=torch.float32)
            sk_tickets = torch.zeros(P // 64, device=dev, dtype=torch.int32)
            sk_ks = max(1, min(8, (H + E) // 832))
        w1t, w2t, w3t, w4t = w1.t(), w2.t(), w3.t(), w4.t()
        # cat_st fused with the NEXT step's reset assembly removes the
        # standalone scan_resets_fwd launch from steps 1..T-1
        fuse_resets = (hasattr(ext, "cat_st_resets_fwd") and H % S == 0
                       and os.environ.get("SHEEPRL_AMD_NO_CATRESETS", "0") != "1")
        # step 0's inputs come from the standalone reset kernel (zero priors)
        ext.scan_resets_fwd(z0, iz, h0, ih, actions[0], f_all[0], x_s[0], hu_s[0], True)
        for t in range(T):
            if t > 0 and not fuse_resets:
                ext.scan_resets_fwd(z_seq[t - 1], iz, h_seq[t - 1], ih, actions[t], f_all[t],
                                    x_s[t], hu_s[t], False)
            if g16:
                torch.mm(x_s[t], w1t, out=g1_s[t])
                ext.ln_act_fwd_o(g1_s[t], lnw1, lnb1, eps, True, hu_s[t, :, H:], mr1_s[0, t], mr1_s[1, t])
                torch.mm(hu_s[t], w2t, out=y_s[t])
                ext.gru_gates_fwd_o(y_s[t], hu_s[t, :, :H], lnwg, lnbg, eps,
                                    h_seq[t], r_s[t, :, :H], mrg_s[0, t], mrg_s[1, t])
                if g16_v3:
                    ext.scan3_f3(r_s[t], w3, lnw3, lnb3, p_s[t], g3_s[t], mr3_s[0, t],
                                 mr3_s[1, t], s3_scr, s3_tk, s3_tk2, s3_ws[t], eps, t + 1)
                else:
                    ext.g16_splitk(r_s[t], w3, None, sk_scratch, sk_tickets, g3_s[t], sk_ks)
                    ext.ln_act_fwd_o(g3_s[t], lnw3, lnb3, eps, True, p_s[t], mr3_s[0, t], mr3_s[1, t])
                torch.addmm(b4, p_s[t], w4t, out=raw)
            else:
                torch.mm(x_s[t], w1t, out=g1_s[t])
                ext.ln_act_fwd_o(g1_s[t], lnw1, lnb1, eps, True, hu_s[t, :, H:], mr1_s[0, t], mr1_s[1, t])
                torch.mm(hu_s[t], w2t, out=y_s[t])
                ext.gru_gates_fwd_o(y_s[t], hu_s[t, :, :H], lnwg, lnbg, eps,
                                    h_seq[t], r_s[t, :, :H], mrg_s[0, t], mrg_s[1, t])
                torch.mm(r_s[t], w3t, out=g3_s[t])
                ext.ln_act_fwd_o(g3_s[t], lnw3, lnb3, eps, True, p_s[t], mr3_s[0, t], mr3_s[1, t])
                torch.addmm(b4, p_s[t], w4t, out=raw)
            if fuse_resets and t + 1 < T:
                ext.cat_st_resets_fwd(raw.view(B, S, discrete), urand_all[t], unimix,
                                      m_seq[t].view(B, S, discrete), z_seq[t], s_s[t],
                                      iz, h_seq[t], ih, actions[t + 1], f_all[t + 1],
                                      x_s[t + 1], hu_s[t + 1])
            else:
                ext.cat_st_fwd_o(raw.view(B, S, discrete), urand_all[t], unimix,
                                 m_seq[t].view(B, S, discrete), z_seq[t].view(B, S, discrete), s_s[t])

        ctx.save_for_backward(
            f_all, w1, lnw1, lnb1, w2, lnwg, lnbg, w3, lnw3, lnb3, w4,
            x_s, g1_s, mr1_s, hu_s, y_s, mrg_s, r_s, g3_s, mr3_s, p_s, s_s,
        )
        ctx.dims = (T, B, E, A, H, SK, D, P, S, discrete)
        ctx.unimix = unimix
        ctx.v2 = False
        return h_seq, z_seq, m_seq

    @staticmethod
    def backward(ctx, g_h_seq: Tensor, g_z_seq: Tensor, g_m_seq: Tensor):
        ext = require_ext()
        (
            f_all, w1, lnw1, lnb1, w2, lnwg, lnbg, w3, lnw3, lnb3, w4,
            x_s, g1_s, mr1_s, hu_s, y_s, mrg_s, r_s, g3_s, mr3_s, p_s, s_s,
        ) = ctx.saved_tensors
        (T, B, E, A, H, SK, D, P, S, discrete) = ctx.dims
        unimix = ctx.unimix
        dev = x_s.device
        dt = x_s.dtype
        g_h_seq = g_h_seq.contiguous()
        g_z_seq = g_z_seq.contiguous()
        g_m_seq = g_m_seq.contiguous()

        # stacked per-step output grads (feed the batched weight-grad GEMMs)
        graw_s = torch.empty(T, B, SK, device=dev, dtype=dt)
        gg3_s = torch.empty(T, B, P, device=dev, dtype=dt)
        gy_s = torch.empty(T, B, 3 * H, device=dev, dtype=dt)
        gg1_s = torch.empty(T, B, D, device=dev, dtype=dt)
        gr_s = torch.empty(T, B, H + E, device=dev, dtype=dt)
        g_actions = torch.empty(T, B, A, device=dev, dtype=dt)
        # fp32 in-kernel accumulators, one zero-fill for all of them
        accs = torch.zeros(2 * D + 6 * H + 2 * P + B * (H + SK), device=dev, dtype=torch.float32)
        off = 0
        def _take(n):
            nonlocal off
            out = accs.narrow(0, off, n)
            off += n
            return out
        glnw1, glnb1 = _take(D), _take(D)
        glnwg, glnbg = _take(3 * H), _take(3 * H)
        glnw3, glnb3 = _take(P), _take(P)
        gih_acc = _take(B * H).view(B, H)
        giz_acc = _take(B * SK).view(B, SK)
        # reused per-step work buffers (incoming-gradient sums happen inside
        # the kernels via their optional extra-source arguments)
        gp = torch.empty(B, P, device=dev, dtype=dt)
        ghp = torch.empty(B, H, device=dev, dtype=dt)
        ghu = torch.empty(B, H + D, device=dev, dtype=dt)
        gx = torch.empty(B, SK + A, device=dev, dtype=dt)
        gh_carry = torch.empty(B, H, device=dev, dtype=dt)   # written at t before read at t-1
        gz_carry = torch.empty(B, SK, device=dev, dtype=dt)

        if getattr(ctx, "v2", False):
            # fused-phase backward: 4 launches/step; the per-WG recompute of
            # the small row-local gradients replaces all cross-WG traffic.
            # Weight transposes once per step so the bwd GEMMs read k-contig.
            w1t = ext.transpose2d(w1)
            w2t = ext.transpose2d(w2)
            w3t = ext.transpose2d(w3)
            w4t = ext.transpose2d(w4)
            if ctx.impl == "v3":
                scr_b4 = torch.zeros(16, P, device=dev, dtype=torch.float32)
                tkb4 = torch.zeros(P // 16, device=dev, dtype=torch.int32)
            for t in range(T - 1, -1, -1):
                zc = gz_carry if t < T - 1 else None
                hc = gh_carry if t < T - 1 else None
                if ctx.impl == "v3":
                    ext.scan3_b4(g_m_seq[t].view(B, SK), g_z_seq[t].view(B, SK), zc,
                                 s_s[t].view(B, SK), w4t, graw_s[t], gp, scr_b4, tkb4,
                                 discrete, unimix, T - t)
                else:
                    ext.scan2_b4(g_m_seq[t].view(B, SK), g_z_seq[t].view(B, SK), zc,
                                 s_s[t].view(B, SK), w4t, graw_s[t], gp, discrete, unimix)
                ext.scan2_b3(gp, g3_s[t], lnw3, lnb3, mr3_s[0, t], mr3_s[1, t], w3t,
                             gg3_s[t], glnw3, glnb3, gr_s[t])
                ext.scan2_b2(g_h_seq[t], hc, gr_s[t, :, :H], y_s[t], hu_s[t, :, :H],
                             lnwg, lnbg, mrg_s[0, t], mrg_s[1, t], w2t, f_all[t],
                             gy_s[t], glnwg, glnbg, gh_carry, gih_acc, ghu)
                ext.scan2_b1(ghu[:, H:], g1_s[t], lnw1, lnb1, mr1_s[0, t], mr1_s[1, t],
                             w1t, f_all[t], gg1_s[t], glnw1, glnb1, gz_carry, giz_acc,
                             g_actions[t])
            return _scan_weight_grads(
                ext, T, B, E, A, H, SK, D, P, dt, dev,
                graw_s, gg3_s, gy_s, gg1_s, gr_s, g_actions,
                p_s, r_s, hu_s, x_s,
                gih_acc, giz_acc, glnw1, glnb1, glnwg, glnbg, glnw3, glnb3,
                lnw1, lnb1, lnwg, lnbg, lnw3, lnb3,
            )

        # cat_st backward fused with the NEXT (already-processed) step's reset
        # backward: the z-carry is computed in-register from gx_{t+1}, and the
        # h-carry/action-grad/init-accumulator writes ride along — removing
        # the standalone scan_resets_bwd launch from steps T-2..1
        fuse_b = (hasattr(ext, "cat_st_resets_bwd") and H % S == 0
                  and os.environ.get("SHEEPRL_AMD_NO_CATRESETS", "0") != "1")
        # interleaved weight-grad GEMMs (experiment, default OFF): issuing a
        # partial [out, chunk*B] x [chunk*B, in] GEMM every `chunk` reverse
        # steps — hoping to read slices still hot in L2/MALL — measured
        # SLOWER than the whole-sequence GEMMs (6.14 -> 6.22/6.30/6.70 ms
        # per graphed scan fwd+bwd at chunk 32/16/8, probes/
        # wgrad_chunk_perf.py): the K=T*B GEMM shape amortizes better than
        # the warmth gain, and the partial sums add launches.
        wchunk = int(os.environ.get("SHEEPRL_AMD_WGRAD_CHUNK", "0"))
        wparts = {k: [] for k in ("w4", "w3", "w2", "w1")} if wchunk > 0 else None
        for t in range(T - 1, -1, -1):
            f = f_all[t]
            hc = gh_carry if t < T - 1 else None
            if fuse_b and t < T - 1:
                ext.cat_st_resets_bwd(g_m_seq[t].view(B, S, discrete),
                                      g_z_seq[t].view(B, S, discrete), s_s[t], unimix,
                                      graw_s[t].view(B, S, discrete), ghu, ghp, gx, f_all[t + 1],
                                      gh_carry, g_actions[t + 1], gih_acc, giz_acc)
            else:
                zc = gz_carry.view(B, S, discrete) if t < T - 1 else None
                ext.cat_st_bwd_o(g_m_seq[t].view(B, S, discrete), g_z_seq[t].view(B, S, discrete), zc,
                                 s_s[t], unimix, graw_s[t].view(B, S, discrete))
            torch.mm(graw_s[t], w4, out=gp)
            ext.ln_act_bwd_acc(gp, g3_s[t], lnw3, lnb3, mr3_s[0, t], mr3_s[1, t], True,
                               gg3_s[t], glnw3, glnb3)
            torch.mm(gg3_s[t], w3, out=gr_s[t])
            ext.gru_gates_bwd_acc(g_h_seq[t], hc, gr_s[t, :, :H], y_s[t], hu_s[t, :, :H], lnwg, lnbg,
                                  mrg_s[0, t], mrg_s[1, t], gy_s[t], ghp, glnwg, glnbg)
            torch.mm(gy_s[t], w2, out=ghu)
            ext.ln_act_bwd_acc(ghu[:, H:], g1_s[t], lnw1, lnb1, mr1_s[0, t], mr1_s[1, t], True,
                               gg1_s[t], glnw1, glnb1)
            torch.mm(gg1_s[t], w1, out=gx)
            # one kernel: carries, action grad, init-state accumulators
            # (fused into the next iteration's cat_st backward when fuse_b,
            # except t=0 whose carries are the init-state path)
            if not fuse_b or t == 0:
                ext.scan_resets_bwd(ghu, ghp, gx, f, gh_carry, gz_carry, g_actions[t], gih_acc, giz_acc)
            if wparts is not None and t % wchunk == 0:
                sl = slice(t, min(t + wchunk, T))
                n = (sl.stop - sl.start) * B
                wparts["w4"].append(torch.mm(graw_s[sl].view(n, SK).t(), p_s[sl].view(n, P)))
                wparts["w3"].append(torch.mm(gg3_s[sl].view(n, P).t(), r_s[sl].view(n, H + E)))
                wparts["w2"].append(torch.mm(gy_s[sl].view(n, 3 * H).t(), hu_s[sl].view(n, H + D)))
                wparts["w1"].append(torch.mm(gg1_s[sl].view(n, D).t(), x_s[sl].view(n, SK + A)))

        gws = None
        if wparts is not None:
            gws = {k: v[0] if len(v) == 1 else torch.stack(v).sum(0) for k, v in wparts.items()}
        return _scan_weight_grads(
            ext, T, B, E, A, H, SK, D, P, dt, dev,
            graw_s, gg3_s, gy_s, gg1_s, gr_s, g_actions,
            p_s, r_s, hu_s, x_s,
            gih_acc, giz_acc, glnw1, glnb1, glnwg, glnbg, glnw3, glnb3,
            lnw1, lnb1, lnwg, lnbg, lnw3, lnb3,
            gws=gws,
        )


def _scan_weight_grads(
    ext, T, B, E, A, H, SK, D, P, dt, dev,
    graw_s, gg3_s, gy_s, gg1_s, gr_s, g_actions,
    p_s, r_s, hu_s, x_s,
    gih_acc, giz_acc, glnw1, glnb1, glnwg, glnbg, glnw3, glnb3,
    lnw1, lnb1, lnwg, lnbg, lnw3, lnb3,
    gws=None,
):
    # batched weight grads: one MFMA GEMM per weight over all T*B rows
    # (or the summed interleaved partials computed inside the reverse loop)
    TB = T * B
    ones_row = torch.ones(1, TB, device=dev, dtype=dt)
    gW4 = gws["w4"] if gws else torch.mm(graw_s.view(TB, SK).t(), p_s.view(TB, P))
    # bias grad as a GEMM against ones: torch's bf16 column-reduce sum(0)
    # was measured at ~300 us for this shape; the GEMV path is ~5 us.
    gb4 = torch.mm(ones_row, graw_s.view(TB, SK)).view(SK)
    gW3 = gws["w3"] if gws else torch.mm(gg3_s.view(TB, P).t(), r_s.view(TB, H + E))
    gW2 = gws["w2"] if gws else torch.mm(gy_s.view(TB, 3 * H).t(), hu_s.view(TB, H + D))
    gW1 = gws["w1"] if gws else torch.mm(gg1_s.view(TB, D).t(), x_s.view(TB, SK + A))
    return (
        gr_s[:, :, H:],                       # g_embed (strided view is fine)
        g_actions, None,
        gih_acc.to(dt).view(1, B, H),
        giz_acc.to(dt).view(1, B, SK),
        gW1, glnw1.to(lnw1.dtype), glnb1.to(lnb1.dtype),
        gW2, glnwg.to(lnwg.dtype), glnbg.to(lnbg.dtype),
        gW3, glnw3.to(lnw3.dtype), glnb3.to(lnb3.dtype),
        gW4, gb4,
        None, None, None, None,
    )


def rssm_scan(
    rssm: Any,
    embed: Tensor,
    actions: Tensor,
    is_first: Tensor,
    initial_states: Tuple[Tensor, Tensor],
    urand: Tensor = None,
) -> Tuple[Tensor, Tensor, Tensor]:
    """Run the coupled RSSM posterior scan with the fused Function.

    Returns (recurrent_states [T,B,H], posteriors_flat [T,B,S*K],
    posteriors_logits fp32 [T,B,S*K]).  Requires the canonical DV3 module
    structure (1-hidden-layer representation MLP, LN+SiLU everywhere).
    """
    mlp_block = rssm.recurrent_model.mlp
    gru = rssm.recurrent_model.rnn
    rep = rssm.representation_model.model
    init_h, init_z = initial_states
    T, B = embed.shape[:2]
    dt = next(rssm.recurrent_model.parameters()).dtype
    return _RSSMScan.apply(
        embed.to(dt).contiguous(),
        actions.to(dt).contiguous(),
        is_first,
        init_h.reshape(1, B, -1).to(dt).contiguous(),
        init_z.reshape(1, B, -1).to(dt).contiguous(),
        mlp_block.linear.weight,
        mlp_block.ln_weight,
        mlp_block.ln_bias,
        gru.linear.weight,
        gru.ln_weight,
        gru.ln_bias,
        rep[0].linear.weight,
        rep[0].ln_weight,
        rep[0].ln_bias,
        rep[1].linear.weight,
        rep[1].linear.bias,
        urand,
        float(rssm.unimix),
        1e-3,
        int(rssm.discrete),
    )


def scan_applicable(rssm: Any) -> bool:
    """True when the module structure matches the fused scan's contract."""
    try:
        mlp_block = rssm.recurrent_model.mlp
        gru = rssm.recurrent_model.rnn
        rep = rssm.representation_model.model
        return (
            mlp_block.layer_norm
            and mlp_block.linear.bias is None
            and gru.linear.bias is None
            and len(rep) == 2
            and rep[0].layer_norm
            and rep[0]._act_name == "silu"
            and not rep[1].layer_norm
            and rep[1].linear.bias is not None
        )
    except AttributeError:
        return False
