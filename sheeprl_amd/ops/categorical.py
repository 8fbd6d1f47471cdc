"""Fused unimix-categorical straight-through head.

Replaces the DV3 stochastic-state subgraph (unimix mixing dreamer_v3/agent.py
:437-449 + one-hot ST sampling dreamer_v2/utils.py:44): softmax -> uniform
mix -> log -> gumbel sample -> one-hot -> straight-through, ~20 eager kernels,
collapsed to rand + one fwd kernel (+ one bwd kernel).

Returns (mixed_log_probs fp32, st_sample input-dtype).  The straight-through
gradient of the sample flows through the mixed probabilities p, identical to
``onehot + p - p.detach()``.
"""

from __future__ import annotations

from typing import Tuple

import torch
from torch import Tensor

from sheeprl_amd.ops._ext import require_ext, use_hip


class _CategoricalST(torch.autograd.Function):
    @staticmethod
    def forward(ctx, raw: Tensor, unimix: float, sample: bool) -> Tuple[Tensor, Tensor]:
        K = raw.shape[-1]
        if use_hip(raw):
            rawc = raw.contiguous()
            u = torch.rand(rawc.shape, dtype=torch.float32, device=raw.device) if sample else None
            m, onehot, s = require_ext().cat_st_fwd(rawc, u, float(unimix), bool(sample))
        else:
            s = torch.softmax(raw.float(), dim=-1)
            p = (1.0 - unimix) * s + unimix / K
            m = torch.log(p)
            if sample:
                u = torch.rand_like(m)
                g = -torch.log(torch.clamp(-torch.log(u.clamp_min(1e-20)), min=1e-20))
                idx = (m + g).argmax(-1)
            else:
                idx = m.argmax(-1)
            onehot = torch.nn.functional.one_hot(idx, K).to(raw.dtype)
        ctx.save_for_backward(s)
        ctx.unimix = unimix
        ctx.raw_dtype = raw.dtype
        return m, onehot

    @staticmethod
    def backward(ctx, gm: Tensor, gon: Tensor):
        (s,) = ctx.saved_tensors
        unimix = ctx.unimix
        if use_hip(s):
            graw = require_ext().cat_st_bwd(gm.contiguous(), gon.contiguous().to(ctx.raw_dtype), s, float(unimix))
        else:
            K = s.shape[-1]
            p = (1.0 - unimix) * s + unimix / K
            t = gm.float() / p + gon.float()
            acc = (t * s).sum(-1, keepdim=True)
            graw = ((1.0 - unimix) * s * (t - acc)).to(ctx.raw_dtype)
        return graw, None, None


def categorical_st(raw_logits: Tensor, unimix: float = 0.01, sample: bool = True) -> Tuple[Tensor, Tensor]:
    """raw_logits [..., K] -> (mixed log-probs fp32 [..., K], one-hot ST sample)."""
    return _CategoricalST.apply(raw_logits, unimix, sample)
