"""SAC losses (parity: sheeprl/algos/sac/loss.py — critic :10, policy :15,
alpha :23)."""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import Tensor


def critic_loss(qs: Tensor, target: Tensor, num_critics: int) -> Tensor:
    return sum(F.mse_loss(qs[..., i : i + 1], target) for i in range(num_critics))


def policy_loss(alpha: Tensor, logp: Tensor, min_q: Tensor) -> Tensor:
    return ((alpha * logp) - min_q).mean()


def entropy_loss(log_alpha: Tensor, logp: Tensor, target_entropy: float) -> Tensor:
    return (-log_alpha * (logp + target_entropy).detach()).mean()
