"""Fused normalization / gate ops (HIP on GPU, PyTorch reference on CPU).

These are the hot elementwise+reduction subgraphs of SURVEY.md §2.8:
* ``layer_norm_act`` — LayerNorm (+optional SiLU) fused in one pass, the
  epilogue of every MLP/CNN block (models.py:16, dreamer_v3/agent.py:42-226).
* ``gru_gates`` — the post-GEMM part of LayerNormGRUCell
  (models.py:396-403): LN over 3H then
  reset=σ(r); cand=tanh(reset*c); update=σ(u-1); h' = update*cand+(1-u')*h.
* ``ema_update_`` — multi-tensor Polyak update (dreamer_v3.py:678-680,
  sac/agent.py:264-267).
"""

from __future__ import annotations

from typing import Iterable, Optional

import torch
from torch import Tensor

from sheeprl_amd.ops._ext import require_ext, use_hip


class _LayerNormAct(torch.autograd.Function):
    """y = act(LN(x) * w + b) over the last dim; act in {none, silu}."""

    @staticmethod
    def forward(ctx, x: Tensor, w: Tensor, b: Tensor, eps: float, act: str) -> Tensor:
        x2 = x.contiguous()
        if use_hip(x):
            y, mean, rstd = require_ext().ln_act_fwd(x2.view(-1, x2.shape[-1]), w, b, float(eps), act == "silu")
            y = y.view_as(x2)
        else:
            xf = x2.float()
            mean = xf.mean(-1, keepdim=True)
            var = xf.var(-1, unbiased=False, keepdim=True)
            rstd = torch.rsqrt(var + eps)
            xhat = (xf - mean) * rstd
            y = xhat * w.float() + b.float()
            if act == "silu":
                y = y * torch.sigmoid(y)
            y = y.to(x.dtype)
            mean = mean.squeeze(-1)
            rstd = rstd.squeeze(-1)
        ctx.save_for_backward(x2, w, b, mean, rstd)
        ctx.act = act
        ctx.eps = eps
        return y

    @staticmethod
    def backward(ctx, gy: Tensor):
        x, w, b, mean, rstd = ctx.saved_tensors
        act = ctx.act
        if use_hip(x):
            D = x.shape[-1]
            gx, gw, gb = require_ext().ln_act_bwd(
                gy.contiguous().view(-1, D), x.view(-1, D), w, b, mean, rstd, act == "silu"
            )
            return gx.view_as(x), gw, gb, None, None
        xf = x.float()
        gyf = gy.float()
        xhat = (xf - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
        z = xhat * w.float() + b.float()
        if act == "silu":
            sig = torch.sigmoid(z)
            dact = sig * (1 + z * (1 - sig))
            gz = gyf * dact
        else:
            gz = gyf
        gw = (gz * xhat).sum(dim=tuple(range(gz.dim() - 1)))
        gb = gz.sum(dim=tuple(range(gz.dim() - 1)))
        gxhat = gz * w.float()
        D = x.shape[-1]
        gx = (gxhat - gxhat.mean(-1, keepdim=True) - xhat * (gxhat * xhat).mean(-1, keepdim=True)) * rstd.unsqueeze(-1)
        return gx.to(x.dtype), gw.to(w.dtype), gb.to(b.dtype), None, None


def layer_norm_act(x: Tensor, weight: Tensor, bias: Tensor, eps: float = 1e-5, act: str = "none") -> Tensor:
    return _LayerNormAct.apply(x, weight, bias, eps, act)


class _GRUGates(torch.autograd.Function):
    """h' from pre-LN projection y=[B,3H] and previous hidden h=[B,H].

    Computes z = LN(y)*w+b; r,c,u = chunk(z,3); r=σ(r); c=tanh(r*c);
    u=σ(u-1); h' = u*c + (1-u)*h.
    """

    @staticmethod
    def forward(ctx, y: Tensor, h: Tensor, w: Tensor, b: Tensor, eps: float) -> Tensor:
        y2, h2 = y.contiguous(), h.contiguous()
        if use_hip(y):
            hnew, mean, rstd = require_ext().gru_gates_fwd(y2, h2, w, b, float(eps))
        else:
            yf = y2.float()
            mean = yf.mean(-1, keepdim=True)
            rstd = torch.rsqrt(yf.var(-1, unbiased=False, keepdim=True) + eps)
            z = ((yf - mean) * rstd) * w.float() + b.float()
            r, c, u = z.chunk(3, -1)
            r = torch.sigmoid(r)
            c = torch.tanh(r * c)
            u = torch.sigmoid(u - 1.0)
            hnew = (u * c + (1 - u) * h2.float()).to(y.dtype)
            mean = mean.squeeze(-1)
            rstd = rstd.squeeze(-1)
        ctx.save_for_backward(y2, h2, w, b, mean, rstd)
        ctx.eps = eps
        return hnew

    @staticmethod
    def backward(ctx, gh: Tensor):
        y, h, w, b, mean, rstd = ctx.saved_tensors
        if use_hip(y):
            gy, ghprev, gw, gb = require_ext().gru_gates_bwd(gh.contiguous(), y, h, w, b, mean, rstd)
            return gy, ghprev, gw, gb, None
        yf, hf, ghf = y.float(), h.float(), gh.float()
        xhat = (yf - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
        z = xhat * w.float() + b.float()
        zr, zc, zu = z.chunk(3, -1)
        r = torch.sigmoid(zr)
        rc = r * zc
        c = torch.tanh(rc)
        u = torch.sigmoid(zu - 1.0)
        # forward: h' = u*c + (1-u)*h
        gu = ghf * (c - hf)
        gc = ghf * u
        ghprev = ghf * (1 - u)
        gzu = gu * u * (1 - u)
        grc = gc * (1 - c * c)
        gzc = grc * r
        gr = grc * zc
        gzr = gr * r * (1 - r)
        gz = torch.cat([gzr, gzc, gzu], dim=-1)
        gw = (gz * xhat).sum(0)
        gb = gz.sum(0)
        gxhat = gz * w.float()
        gy = (gxhat - gxhat.mean(-1, keepdim=True) - xhat * (gxhat * xhat).mean(-1, keepdim=True)) * rstd.unsqueeze(-1)
        return gy.to(y.dtype), ghprev.to(h.dtype), gw.to(w.dtype), gb.to(b.dtype), None


def gru_gates(y: Tensor, h: Tensor, weight: Tensor, bias: Tensor, eps: float = 1e-3) -> Tensor:
    return _GRUGates.apply(y, h, weight, bias, eps)


@torch.no_grad()
def ema_update_(target_params: Iterable[Tensor], source_params: Iterable[Tensor], tau: float) -> None:
    """t <- tau * p + (1 - tau) * t, multi-tensor."""
    tgt = list(target_params)
    src = list(source_params)
    if (
        tgt
        and use_hip(tgt[0])
        # the HIP kernel walks data_ptr flat: both sides must share the same
        # dense layout (true for deepcopy/convert pairs; guard the rest)
        and all(t.stride() == s.stride() for t, s in zip(tgt, src))
    ):
        require_ext().ema_update(tgt, src, float(tau))
        return
    torch._foreach_mul_(tgt, 1.0 - tau)
    torch._foreach_add_(tgt, src, alpha=tau)


@torch.no_grad()
def normalize_obs(obs: Tensor) -> Tensor:
    """uint8 image -> float in [-0.5, 0.5] (parity: dreamer_v3/utils.py:80-91)."""
    if obs.dtype == torch.uint8 and use_hip(obs):
        return require_ext().obs_norm(obs.contiguous())
    return obs.float() / 255.0 - 0.5


class _MaskedLerp(torch.autograd.Function):
    """y = (1-f)*x + f*init with a per-row mask f (episode-reset masking in
    the RSSM scan; one kernel instead of ~4 eager broadcast ops)."""

    @staticmethod
    def forward(ctx, x: Tensor, init: Optional[Tensor], f: Tensor) -> Tensor:
        fc = f.reshape(-1).to(x.dtype).contiguous()  # one mask entry per row
        if use_hip(x):
            initc = init.contiguous().to(x.dtype) if init is not None else None
            y = require_ext().masked_lerp_fwd(x.contiguous(), initc, fc)
        else:
            fb = fc.view(*x.shape[:-1], 1)
            y = (1 - fb) * x + (fb * init if init is not None else 0.0)
        ctx.save_for_backward(fc)
        ctx.has_init = init is not None
        return y

    @staticmethod
    def backward(ctx, g: Tensor):
        (fc,) = ctx.saved_tensors
        if use_hip(g):
            gx, ginit = require_ext().masked_lerp_bwd(g.contiguous(), fc.to(g.dtype), ctx.has_init)
            return gx, (ginit if ctx.has_init else None), None
        fb = fc.view(*g.shape[:-1], 1).to(g.dtype)
        gx = (1 - fb) * g
        ginit = fb * g if ctx.has_init else None
        return gx, ginit, None


def masked_lerp(x: Tensor, init: Optional[Tensor], f: Tensor) -> Tensor:
    """Rows of ``x``/``init`` are lerped by the per-row mask ``f`` ([*, 1])."""
    return _MaskedLerp.apply(x, init, f)


class _TwoHotLogProb(torch.autograd.Function):
    """Fused two-hot cross-entropy: logsumexp + uniform-bin two-hot of
    symlog(value) + inner product in one kernel each way (the torch chain is
    ~12 launches; reference distribution.py:224-276 semantics)."""

    @staticmethod
    def forward(ctx, logits: Tensor, value: Tensor, low: float, high: float) -> Tensor:
        ext = require_ext()
        out, lse = ext.twohot_lp_fwd(logits.contiguous(), value.contiguous(), low, high)
        ctx.save_for_backward(logits, value, lse)
        ctx.lh = (low, high)
        return out.view(logits.shape[:-1])

    @staticmethod
    def backward(ctx, g: Tensor):
        ext = require_ext()
        logits, value, lse = ctx.saved_tensors
        low, high = ctx.lh
        gl = ext.twohot_lp_bwd(g.contiguous().view(-1), logits, value, lse, low, high)
        return gl.view(logits.shape), None, None, None


def twohot_log_prob(logits: Tensor, value: Tensor, low: float = -20.0, high: float = 20.0) -> Tensor:
    """log_prob of a TwoHotEncodingDistribution over a uniform symlog support.

    ``logits`` [..., K] fp32, ``value`` [...] fp32 (no grad); returns [...].
    """
    return _TwoHotLogProb.apply(logits, value.float(), low, high)


class _KLBalanced(torch.autograd.Function):
    """Fused two-sided KL balancing (DV3 world-model loss, reference
    loss.py:64-75): KL(sg(post)||prior) and KL(post||sg(prior)) share the
    forward value; the two outputs route their gradients to prior and post
    respectively."""

    @staticmethod
    def forward(ctx, post_logits: Tensor, prior_logits: Tensor):
        ext = require_ext()
        post = post_logits.contiguous()
        prior = prior_logits.contiguous()
        kl, kls = ext.klbal_fwd(post, prior)
        ctx.save_for_backward(post, prior, kls)
        out = kl.view(post_logits.shape[:-2])
        return out, out.clone()

    @staticmethod
    def backward(ctx, g_dyn: Tensor, g_rep: Tensor):
        ext = require_ext()
        post, prior, kls = ctx.saved_tensors
        g_post, g_prior = ext.klbal_bwd(g_dyn.contiguous().view(-1), g_rep.contiguous().view(-1),
                                        post, prior, kls)
        return g_post, g_prior


def kl_balanced(post_logits: Tensor, prior_logits: Tensor):
    """Returns (kl_dynamic, kl_representation): numerically the same
    KL(post||prior) summed over the stoch dim, but the first backpropagates
    only into ``prior_logits`` and the second only into ``post_logits``."""
    return _KLBalanced.apply(post_logits, prior_logits)
