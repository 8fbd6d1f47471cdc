"""Fused RSSM sequence scan (forward + hand-written backward).

The DV3 dynamic-learning scan is T=64 sequential steps; under autograd each
step leaves ~40 bookkeeping kernels in the backward (per-use gradient
accumulation adds on every weight, cat-backward copies, zero fills).  This
Function runs the whole scan with explicit kernels and accumulates weight
gradients in-place with `addmm_` — one GEMM per weight per step, nothing else.

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

import torch
from torch import Tensor

from sheeprl_amd.ops._ext import require_ext


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
        # saved intermediates
        x_s = torch.empty(T, B, SK + A, device=dev, dtype=dt)
        g1_s = torch.empty(T, B, D, device=dev, dtype=dt)
        m1_s = torch.empty(T, B, device=dev, dtype=torch.float32)
        r1_s = torch.empty(T, B, device=dev, dtype=torch.float32)
        hin_s = torch.empty(T, B, H, device=dev, dtype=dt)
        hu_s = torch.empty(T, B, H + D, device=dev, dtype=dt)
        y_s = torch.empty(T, B, 3 * H, device=dev, dtype=dt)
        mg_s = torch.empty(T, B, device=dev, dtype=torch.float32)
        rg_s = torch.empty(T, B, device=dev, dtype=torch.float32)
        g3_s = torch.empty(T, B, P, device=dev, dtype=dt)
        m3_s = torch.empty(T, B, device=dev, dtype=torch.float32)
        r3_s = torch.empty(T, B, device=dev, dtype=torch.float32)
        p_s = torch.empty(T, B, P, device=dev, dtype=dt)
        s_s = torch.empty(T, B, S, discrete, device=dev, dtype=torch.float32)

        ih = init_h[0]
        iz = init_z[0]
        h = torch.zeros(B, H, device=dev, dtype=dt)
        z = torch.zeros(B, SK, device=dev, dtype=dt)
        for t in range(T):
            f = is_first[t].to(dt)
            a_eff = ext.masked_lerp_fwd(actions[t].contiguous().to(dt), None, f.reshape(-1))
            h_in = ext.masked_lerp_fwd(h, ih, f.reshape(-1))
            z_in = ext.masked_lerp_fwd(z, iz, f.reshape(-1))
            x = torch.cat((z_in, a_eff), dim=-1)
            g1 = x @ w1.t()
            u, m1, r1 = ext.ln_act_fwd(g1, lnw1, lnb1, eps, True)
            hu = torch.cat((h_in, u), dim=-1)
            y = hu @ w2.t()
            h, mg, rg = ext.gru_gates_fwd(y, h_in, lnwg, lnbg, eps)
            r = torch.cat((h, embed[t]), dim=-1)
            g3 = r @ w3.t()
            p, m3, r3 = ext.ln_act_fwd(g3, lnw3, lnb3, eps, True)
            raw = torch.addmm(b4, p, w4.t())
            urand = torch.rand(B, S, discrete, device=dev, dtype=torch.float32)
            m, onehot, s = ext.cat_st_fwd(raw.view(B, S, discrete), urand, unimix, True)
            z = onehot.view(B, SK).to(dt)

            h_seq[t] = h
            z_seq[t] = z
            m_seq[t] = m.view(B, SK)
            x_s[t] = x
            g1_s[t] = g1
            m1_s[t], r1_s[t] = m1, r1
            hin_s[t] = h_in
            hu_s[t] = hu
            y_s[t] = y
            mg_s[t], rg_s[t] = mg, rg
            g3_s[t] = g3
            m3_s[t], r3_s[t] = m3, r3
            p_s[t] = p
            s_s[t] = s

        ctx.save_for_backward(
            embed, actions, is_first, init_h, init_z,
            w1, lnw1, lnb1, w2, lnwg, lnbg, w3, lnw3, lnb3, w4, b4,
            h_seq, z_seq, x_s, g1_s, m1_s, r1_s, hin_s, hu_s, y_s, mg_s, rg_s,
            g3_s, m3_s, r3_s, p_s, s_s,
        )
        ctx.dims = (T, B, E, A, H, SK, D, P, S, discrete)
        ctx.unimix = unimix
        ctx.eps = eps
        return h_seq, z_seq, m_seq

    @staticmethod
    def backward(ctx, g_h_seq: Tensor, g_z_seq: Tensor, g_m_seq: Tensor):
        ext = require_ext()
        (
            embed, actions, is_first, init_h, init_z,
            w1, lnw1, lnb1, w2, lnwg, lnbg, w3, lnw3, lnb3, w4, b4,
            h_seq, z_seq, x_s, g1_s, m1_s, r1_s, hin_s, hu_s, y_s, mg_s, rg_s,
            g3_s, m3_s, r3_s, p_s, s_s,
        ) = ctx.saved_tensors
        (T, B, E, A, H, SK, D, P, S, discrete) = ctx.dims
        unimix, eps = ctx.unimix, ctx.eps
        dt = embed.dtype

        gW1 = torch.zeros_like(w1)
        glnw1 = torch.zeros_like(lnw1)
        glnb1 = torch.zeros_like(lnb1)
        gW2 = torch.zeros_like(w2)
        glnwg = torch.zeros_like(lnwg)
        glnbg = torch.zeros_like(lnbg)
        gW3 = torch.zeros_like(w3)
        glnw3 = torch.zeros_like(lnw3)
        glnb3 = torch.zeros_like(lnb3)
        gW4 = torch.zeros_like(w4)
        gb4 = torch.zeros_like(b4)
        g_embed = torch.empty_like(embed)
        g_actions = torch.empty_like(actions)
        g_init_h = torch.zeros_like(init_h)
        g_init_z = torch.zeros_like(init_z)

        gh_carry = torch.zeros(B, H, device=embed.device, dtype=dt)
        gz_carry = torch.zeros(B, SK, device=embed.device, dtype=dt)
        for t in range(T - 1, -1, -1):
            f = is_first[t].to(dt).reshape(-1)
            gm_t = g_m_seq[t].contiguous().view(B, S, discrete)
            gz_t = (g_z_seq[t] + gz_carry).view(B, S, discrete)
            graw = ext.cat_st_bwd(gm_t, gz_t.to(dt), s_s[t], unimix).view(B, SK)
            # W4 / b4
            gp = graw @ w4
            gW4.addmm_(graw.t(), p_s[t])
            gb4.add_(graw.sum(0))
            # ln3
            gg3, gw3_, gb3_ = ext.ln_act_bwd(gp.contiguous(), g3_s[t], lnw3, lnb3, m3_s[t], r3_s[t], True)
            glnw3.add_(gw3_)
            glnb3.add_(gb3_)
            # W3 over r = [h_t, embed_t]
            gr = gg3 @ w3
            r = torch.cat((h_seq[t], embed[t]), dim=-1)
            gW3.addmm_(gg3.t(), r)
            gh_total = g_h_seq[t] + gh_carry + gr[:, :H]
            g_embed[t] = gr[:, H:]
            # GRU gates
            gy, gh_in1, gwg_, gbg_ = ext.gru_gates_bwd(
                gh_total.contiguous(), y_s[t], hin_s[t], lnwg, lnbg, mg_s[t], rg_s[t]
            )
            glnwg.add_(gwg_)
            glnbg.add_(gbg_)
            # W2 over hu = [h', u]
            ghu = gy @ w2
            gW2.addmm_(gy.t(), hu_s[t])
            gh_in = ghu[:, :H] + gh_in1
            gu = ghu[:, H:]
            # ln1
            gg1, gw1_, gb1_ = ext.ln_act_bwd(gu.contiguous(), g1_s[t], lnw1, lnb1, m1_s[t], r1_s[t], True)
            glnw1.add_(gw1_)
            glnb1.add_(gb1_)
            # W1 over x = [z', a']
            gx = gg1 @ w1
            gW1.addmm_(gg1.t(), x_s[t])
            gz_in = gx[:, :SK]
            ga_eff = gx[:, SK:]
            # masked reset backward (carries flow to step t-1)
            gh_carry, gih = ext.masked_lerp_bwd(gh_in.contiguous(), f, True)
            g_init_h[0].add_(gih)
            gz_carry, giz = ext.masked_lerp_bwd(gz_in.contiguous(), f, True)
            g_init_z[0].add_(giz)
            ga, _ = ext.masked_lerp_bwd(ga_eff.contiguous(), f, False)
            g_actions[t] = ga

        return (
            g_embed, g_actions, None, g_init_h, g_init_z,
            gW1, glnw1, glnb1, gW2, glnwg, glnbg, gW3, glnw3, glnb3, gW4, gb4,
            None, None, None,
        )


def rssm_scan(
    rssm: Any,
    embed: Tensor,
    actions: Tensor,
    is_first: Tensor,
    initial_states: Tuple[Tensor, Tensor],
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
        embed.to(dt),
        actions.to(dt),
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
