"""Fused LSTM sequence scan for ppo_recurrent.

The reference steps ``nn.LSTM`` per timestep in a Python loop with in-loop
``.contiguous()`` calls (sheeprl/algos/ppo_recurrent/agent.py:39-43 + the
packed-sequence handling) — the same launch-storm shape the DV3 scan work
eliminated.  Here the input projection is ONE batched GEMM over [T*B], each
step is one ``addmm`` (h-side GEMM) + one fused gates kernel (torch gate
order i|f|g|o with the episode-reset mask folded in), and the hand-written
backward accumulates all weight gradients as three batched GEMMs after the
reverse loop.
"""

from __future__ import annotations

from typing import Tuple

import torch
from torch import Tensor

from sheeprl_amd.ops._ext import require_ext, use_hip


class _LSTMScan(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: Tensor, first: Tensor, h0: Tensor, c0: Tensor,
                w_ih: Tensor, w_hh: Tensor, b: Tensor):
        ext = require_ext()
        T, B, F = x.shape
        H = h0.shape[-1]
        dev, dt = x.device, x.dtype
        xc = x.contiguous()
        gx = torch.addmm(b, xc.reshape(T * B, F), w_ih.t()).view(T, B, 4 * H)
        y_s = torch.empty(T, B, 4 * H, device=dev, dtype=dt)
        h_seq = torch.empty(T, B, H, device=dev, dtype=dt)
        c_seq = torch.empty(T, B, H, device=dev, dtype=dt)
        hprev_s = torch.empty(T, B, H, device=dev, dtype=dt)  # masked h' (GEMM inputs)
        w_hh_t = w_hh.t()
        h, c = h0.contiguous(), c0.contiguous()
        first = first.reshape(T, B).to(dt).contiguous()
        for t in range(T):
            hp = hprev_s[t]
            torch.mul(h, (1.0 - first[t]).unsqueeze(-1), out=hp)
            torch.addmm(gx[t], hp, w_hh_t, out=y_s[t])
            ext.lstm_gates_fwd(y_s[t], c, first[t], h_seq[t], c_seq[t])
            h, c = h_seq[t], c_seq[t]
        ctx.save_for_backward(xc, first, c0, y_s, c_seq, hprev_s, w_ih, w_hh)
        return h_seq, h_seq[-1].clone(), c_seq[-1].clone()

    @staticmethod
    def backward(ctx, g_hseq: Tensor, g_hlast: Tensor, g_clast: Tensor):
        ext = require_ext()
        x, first, c0, y_s, c_seq, hprev_s, w_ih, w_hh = ctx.saved_tensors
        T, B, H4 = y_s.shape
        H = H4 // 4
        F = x.shape[-1]
        dev, dt = x.device, x.dtype
        g_hseq = g_hseq.contiguous()
        gy_s = torch.empty(T, B, 4 * H, device=dev, dtype=dt)
        gh_raw = torch.empty(B, H, device=dev, dtype=dt)   # gy_{t+1} @ w_hh
        gc_carry = g_clast.contiguous().clone()
        gh2 = None
        first2 = None
        gh_t = g_hseq[T - 1] + g_hlast
        for t in range(T - 1, -1, -1):
            c_prev = c_seq[t - 1] if t > 0 else c0
            ext.lstm_gates_bwd(gh_t.contiguous(), gh2, first2, gc_carry, y_s[t],
                               c_prev.contiguous(), c_seq[t], first[t], gy_s[t], gc_carry)
            if t > 0:
                torch.mm(gy_s[t], w_hh, out=gh_raw)
                gh2 = gh_raw
                first2 = first[t]
                gh_t = g_hseq[t - 1]
        # grads to the initial states through step 0
        gh0 = torch.mm(gy_s[0], w_hh) * (1.0 - first[0]).unsqueeze(-1)
        gc0 = gc_carry
        TB = T * B
        gx_flat = gy_s.view(TB, 4 * H).mm(w_ih)
        gW_ih = gy_s.view(TB, 4 * H).t().mm(x.view(TB, F))
        gW_hh = gy_s.view(TB, 4 * H).t().mm(hprev_s.view(TB, H))
        gb = torch.ones(1, TB, device=dev, dtype=dt).mm(gy_s.view(TB, 4 * H)).view(4 * H)
        return gx_flat.view(T, B, F), None, gh0, gc0, gW_ih, gW_hh, gb


def lstm_scan(x: Tensor, is_first: Tensor, states: Tuple[Tensor, Tensor], lstm: torch.nn.LSTM):
    """Run the masked LSTM scan with the fused kernels; returns
    (out [T,B,H], (h_last [1,B,H], c_last [1,B,H])).

    Semantics match stepping ``lstm`` per t with h/c zeroed where
    ``is_first[t] == 1`` (the reference's packed-sequence episode handling).
    """
    h0, c0 = states
    b = lstm.bias_ih_l0 + lstm.bias_hh_l0
    out, h_last, c_last = _LSTMScan.apply(
        x, is_first, h0.reshape(-1, h0.shape[-1]), c0.reshape(-1, c0.shape[-1]),
        lstm.weight_ih_l0, lstm.weight_hh_l0, b,
    )
    return out, (h_last.unsqueeze(0), c_last.unsqueeze(0))


def lstm_scan_applicable(lstm: torch.nn.LSTM, x: Tensor) -> bool:
    return (
        use_hip(x)
        and lstm.num_layers == 1
        and not lstm.bidirectional
        and getattr(lstm, "bias_ih_l0", None) is not None
        and not lstm.batch_first
    )
