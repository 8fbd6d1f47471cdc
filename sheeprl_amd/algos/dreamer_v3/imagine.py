"""Launch-lean imagination rollout for the discrete (REINFORCE) DV3 actor.

In the reference the behaviour-learning rollout (dreamer_v3.py:235-241) runs
the actor and ``rssm.imagination`` as module calls — ~35 kernel launches per
imagined step, all on a latency floor.  For DISCRETE actions nothing
backpropagates through the rollout (the actor loss recomputes its policies on
``imagined_trajectories.detach()`` and the advantage is detached), so the
rollout is pure inference: this version runs it under ``no_grad`` with
preallocated stacked buffers, strided-out kernels, and batched philox — ~17
launches per step and zero autograd bookkeeping.

Semantics are identical to the module loop (same kernels, same philox
consumption ORDER differs only in that the per-step ``torch.rand`` calls are
replaced by two batched draws — any iid uniforms are equivalent here);
validated against the module loop in tests/test_gpu_kernels.py.
"""

from __future__ import annotations

from typing import Any, Tuple

import torch
from torch import Tensor

from sheeprl_amd.ops._ext import require_ext


def imagine_applicable(rssm: Any, actor: Any) -> bool:
    """Fast rollout needs the canonical DV3 discrete setup: single discrete
    actor head, LN+SiLU dense blocks without bias everywhere, and the
    standard transition-model shape."""
    try:
        if actor.is_continuous or len(actor.mlp_heads) != 1:
            return False
        trans = rssm.transition_model.model
        mlp_block = rssm.recurrent_model.mlp
        gru = rssm.recurrent_model.rnn
        actor_blocks = list(actor.model.model)
        return (
            mlp_block.layer_norm
            and mlp_block.linear.bias is None
            and gru.linear.bias is None
            and len(trans) == 2
            and trans[0].layer_norm
            and trans[0]._act_name == "silu"
            and not trans[1].layer_norm
            and trans[1].linear.bias is not None
            and len(actor_blocks) >= 1
            and all(b.layer_norm and b._act_name == "silu" and b.linear.bias is None for b in actor_blocks)
        )
    except AttributeError:
        return False


@torch.no_grad()
def imagine_rollout(
    rssm: Any,
    actor: Any,
    z0: Tensor,  # [B, SK] detached posteriors
    h0: Tensor,  # [B, H] detached recurrent states
    horizon: int,
    urand_t: Tensor = None,  # [HZ, B, S, K] test parity hook
    urand_a: Tensor = None,  # [HZ+1, B, A] test parity hook
) -> Tuple[Tensor, Tensor]:
    """Returns (imagined_trajectories [horizon+1, B, SK+H],
    imagined_actions [horizon+1, B, A])."""
    ext = require_ext()
    mlp_block = rssm.recurrent_model.mlp
    gru = rssm.recurrent_model.rnn
    trans = rssm.transition_model.model
    head = actor.mlp_heads[0]
    actor_blocks = list(actor.model.model)
    eps = 1e-3
    unimix = float(rssm.unimix)
    a_unimix = float(actor._unimix)
    K = int(rssm.discrete)

    dt = z0.dtype
    dev = z0.device
    B, SK = z0.shape
    H = h0.shape[-1]
    S = SK // K
    A = head.out_features
    DU = mlp_block.linear.out_features
    P = trans[0].linear.out_features
    HZ = horizon

    w1 = mlp_block.linear.weight
    w2 = gru.linear.weight
    wt1 = trans[0].linear.weight
    wt2, bt2 = trans[1].linear.weight, trans[1].linear.bias

    traj = torch.empty(HZ + 1, B, SK + H, device=dev, dtype=dt)
    acts = torch.empty(HZ + 1, B, A, device=dev, dtype=dt)
    if urand_t is None:
        urand_t = torch.rand(HZ, B, S, K, device=dev, dtype=torch.float32)
    if urand_a is None:
        urand_a = torch.rand(HZ + 1, B, A, device=dev, dtype=torch.float32)
    # reused scratch
    xs = torch.empty(B, SK + A, device=dev, dtype=dt)
    g1 = torch.empty(B, DU, device=dev, dtype=dt)
    hu = torch.empty(B, H + DU, device=dev, dtype=dt)
    y = torch.empty(B, 3 * H, device=dev, dtype=dt)
    h_buf = torch.empty(B, H, device=dev, dtype=dt)
    tg = torch.empty(B, P, device=dev, dtype=dt)
    tp = torch.empty(B, P, device=dev, dtype=dt)
    traw = torch.empty(B, SK, device=dev, dtype=dt)
    mr = torch.empty(2, B, device=dev, dtype=torch.float32)
    m_tmp = torch.empty(B, SK, device=dev, dtype=torch.float32)
    s_tmp = torch.empty(B, S, K, device=dev, dtype=torch.float32)
    am_tmp = torch.empty(B, A, device=dev, dtype=torch.float32)
    as_tmp = torch.empty(B, A, device=dev, dtype=torch.float32)
    a_hidden = [torch.empty(B, b.linear.out_features, device=dev, dtype=dt) for b in actor_blocks]
    a_pre = [torch.empty(B, b.linear.out_features, device=dev, dtype=dt) for b in actor_blocks]
    alogits = torch.empty(B, A, device=dev, dtype=dt)

    def actor_step(i: int) -> None:
        x = traj[i]
        for blk, pre, hid in zip(actor_blocks, a_pre, a_hidden):
            torch.mm(x, blk.linear.weight.t(), out=pre)
            ext.ln_act_fwd_o(pre, blk.ln_weight, blk.ln_bias, blk.ln_eps, True, hid, mr[0], mr[1])
            x = hid
        torch.addmm(head.bias, x, head.weight.t(), out=alogits)
        # the sampled action ALSO lands in the next step's GEMM input slice
        ext.cat_st_fwd_o(alogits, urand_a[i], a_unimix, am_tmp, acts[i], as_tmp, xs[:, SK:])

    # the producing kernels write every next-step input in place (second /
    # third strided outputs), so the per-step assembly copies are gone; only
    # the i=1 inputs (z0/h0, not produced by a kernel here) are staged
    traj[0, :, :SK] = z0
    traj[0, :, SK:] = h0
    h_buf.copy_(h0)
    xs[:, :SK].copy_(z0)
    hu[:, :H].copy_(h0)
    actor_step(0)
    for i in range(1, HZ + 1):
        # --- RSSM imagination step (recurrent_model + transition) ---
        torch.mm(xs, w1.t(), out=g1)
        ext.ln_act_fwd_o(g1, mlp_block.ln_weight, mlp_block.ln_bias, eps, True, hu[:, H:], mr[0], mr[1])
        torch.mm(hu, w2.t(), out=y)
        # h_i goes to h_buf, the trajectory slice, AND next step's hu[:, :H]
        # (the kernel reads each hu row element before overwriting it)
        ext.gru_gates_fwd_o(y, hu[:, :H], gru.ln_weight, gru.ln_bias, eps, h_buf, traj[i][:, SK:],
                            mr[0], mr[1], hu[:, :H])
        torch.mm(h_buf, wt1.t(), out=tg)
        ext.ln_act_fwd_o(tg, trans[0].ln_weight, trans[0].ln_bias, trans[0].ln_eps, True, tp, mr[0], mr[1])
        torch.addmm(bt2, tp, wt2.t(), out=traw)
        ext.cat_st_fwd_o(traw.view(B, S, K), urand_t[i - 1], unimix, m_tmp.view(B, S, K),
                         traj[i][:, :SK], s_tmp, xs[:, :SK])
        # --- policy on the new latent ---
        actor_step(i)
    return traj, acts
