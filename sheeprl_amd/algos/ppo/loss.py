"""PPO losses (parity: sheeprl/algos/ppo/loss.py — policy_loss :6,
value_loss :45, entropy_loss :65)."""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import Tensor


def policy_loss(
    new_logprobs: Tensor,
    logprobs: Tensor,
    advantages: Tensor,
    clip_coef: float,
    reduction: str = "mean",
) -> Tensor:
    logratio = new_logprobs - logprobs
    ratio = logratio.exp()
    pg_loss1 = -advantages * ratio
    pg_loss2 = -advantages * torch.clamp(ratio, 1 - clip_coef, 1 + clip_coef)
    loss = torch.max(pg_loss1, pg_loss2)
    if reduction == "mean":
        return loss.mean()
    if reduction == "sum":
        return loss.sum()
    return loss


def value_loss(
    new_values: Tensor,
    old_values: Tensor,
    returns: Tensor,
    clip_coef: float,
    clip_vloss: bool,
    reduction: str = "mean",
) -> Tensor:
    if not clip_vloss:
        return F.mse_loss(new_values, returns, reduction=reduction)
    v_loss_unclipped = (new_values - returns) ** 2
    v_clipped = old_values + torch.clamp(new_values - old_values, -clip_coef, clip_coef)
    v_loss_clipped = (v_clipped - returns) ** 2
    # reference scales the clipped branch by 0.5 (sheeprl/algos/ppo/loss.py:61)
    v_loss = 0.5 * torch.max(v_loss_unclipped, v_loss_clipped)
    if reduction == "mean":
        return v_loss.mean()
    if reduction == "sum":
        return v_loss.sum()
    return v_loss


def entropy_loss(entropy: Tensor, reduction: str = "mean") -> Tensor:
    if reduction == "mean":
        return -entropy.mean()
    if reduction == "sum":
        return -entropy.sum()
    return -entropy


def ppo_losses(
    new_logprobs: Tensor,
    old_logprobs: Tensor,
    advantages: Tensor,
    new_values: Tensor,
    old_values: Tensor,
    returns: Tensor,
    entropy: Tensor,
    clip_coef: float,
    clip_vloss: bool,
    reduction: str = "mean",
):
    """(policy_loss, value_loss, entropy_loss) triple — ONE fused kernel each
    way on GPU (ops.ppo_losses), the eager composition above elsewhere.
    Value- and gradient-identical by construction (tested in
    tests/test_gpu_kernels.py::test_ppo_losses_fused_matches_eager)."""
    from sheeprl_amd import ops as _ops

    if reduction in ("mean", "sum") and new_logprobs.is_cuda and _ops.use_hip(new_logprobs):
        return _ops.ppo_losses(
            new_logprobs, old_logprobs, advantages, new_values, old_values, returns,
            entropy, clip_coef, clip_vloss, reduction,
        )
    return (
        policy_loss(new_logprobs, old_logprobs, advantages, clip_coef, reduction),
        value_loss(new_values, old_values, returns, clip_coef, clip_vloss, reduction),
        entropy_loss(entropy, reduction),
    )
