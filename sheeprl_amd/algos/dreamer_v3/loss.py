"""Dreamer-V3 world-model loss (parity: sheeprl/algos/dreamer_v3/loss.py:9 —
two-sided KL balancing with free nats :64-75)."""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch
from torch import Tensor


def categorical_kl(p_logits: Tensor, q_logits: Tensor) -> Tensor:
    """KL(P || Q) for [*, stoch, discrete] categorical logits, summed over the
    stoch dimension; computed in fp32 from log-softmaxes (stable under
    bf16-true, SURVEY.md §7 hard-part 2)."""
    p_log = torch.log_softmax(p_logits.float(), dim=-1)
    q_log = torch.log_softmax(q_logits.float(), dim=-1)
    p = p_log.exp()
    return (p * (p_log - q_log)).sum(dim=(-2, -1))


def reconstruction_loss(
    po: Dict[str, object],
    observations: Dict[str, Tensor],
    pr: object,
    rewards: Tensor,
    priors_logits: Tensor,
    posteriors_logits: Tensor,
    kl_dynamic: float = 0.5,
    kl_representation: float = 0.1,
    kl_free_nats: float = 1.0,
    kl_regularizer: float = 1.0,
    pc: Optional[object] = None,
    continue_targets: Optional[Tensor] = None,
    continue_scale_factor: float = 1.0,
) -> Tuple[Tensor, Tensor, Tensor, Tensor, Tensor, Tensor]:
    observation_loss = -sum(po[k].log_prob(observations[k]) for k in po.keys())
    reward_loss = -pr.log_prob(rewards)
    # KL balancing: dynamic = KL(sg(post) || prior), representation = KL(post || sg(prior))
    from sheeprl_amd import ops as _ops

    if posteriors_logits.is_cuda and _ops.use_hip(posteriors_logits):
        kl_dyn_v, kl_rep_v = _ops.kl_balanced(posteriors_logits.float(), priors_logits.float())
        kl = dyn_loss = kl_dyn_v
        free_nats = torch.full_like(dyn_loss, kl_free_nats)
        dyn_loss = kl_dynamic * torch.maximum(dyn_loss, free_nats)
        repr_loss = kl_representation * torch.maximum(kl_rep_v, free_nats)
    else:
        kl = dyn_loss = categorical_kl(posteriors_logits.detach(), priors_logits)
        free_nats = torch.full_like(dyn_loss, kl_free_nats)
        dyn_loss = kl_dynamic * torch.maximum(dyn_loss, free_nats)
        repr_loss = categorical_kl(posteriors_logits, priors_logits.detach())
        repr_loss = kl_representation * torch.maximum(repr_loss, free_nats)
    kl_loss = dyn_loss + repr_loss
    if pc is not None and continue_targets is not None:
        logits = getattr(getattr(pc, "base_dist", None), "logits", None)
        if (
            logits is not None
            and logits.is_cuda
            and logits.shape == continue_targets.shape
            and _ops.use_hip(logits)
        ):
            # fused Bernoulli NLL (one kernel each way)
            continue_loss = continue_scale_factor * -_ops.bernoulli_log_prob(logits, continue_targets, 1)
        else:
            continue_loss = continue_scale_factor * -pc.log_prob(continue_targets)
    else:
        continue_loss = torch.zeros_like(reward_loss)
    rec_loss = (kl_regularizer * kl_loss + observation_loss + reward_loss + continue_loss).mean()
    return rec_loss, kl, kl_loss.mean(), reward_loss.mean(), observation_loss.mean(), continue_loss.mean()


def reconstruction_loss_forked(
    br,
    world_model,
    latent_states: Tensor,
    recurrent_states: Tensor,
    batch_obs: Dict[str, Tensor],
    rewards: Tensor,
    continue_targets: Tensor,
    posteriors_logits: Tensor,
    priors_logits: Optional[Tensor],
    stochastic_size: int,
    discrete_size: int,
    cnn_keys,
    mlp_keys,
    kl_dynamic: float,
    kl_representation: float,
    kl_free_nats: float,
    kl_regularizer: float,
    continue_scale_factor: float,
    no_cast: bool,
):
    """Same math as :func:`reconstruction_loss`, restructured so the four
    independent world-model head chains (decoder, reward, continue, prior+KL)
    each run — forward and, via autograd stream semantics, backward — inside
    their own ``br.fork()`` (see ``parallel/streams.py``).  With ``br``
    disabled the forks are no-ops and this is sequentially identical to the
    original (unit-tested in tests/test_algos/test_losses.py).

    ``priors_logits=None`` computes the batched transition head inside the
    prior fork (the standard-RSSM case where it is off the scan's critical
    path); the decoupled-RSSM path passes its in-loop priors instead.

    Returns the 6 loss terms plus the [*, stoch, discrete]-viewed prior and
    posterior logits (train() reuses them for the entropy metrics).
    """
    import torch.distributions as td

    from sheeprl_amd import ops as _ops
    from sheeprl_amd.distributions.dists import (
        BernoulliSafeMode,
        MSEDistribution,
        SymlogDistribution,
        TwoHotEncodingDistribution,
    )

    _c = (lambda t: t) if no_cast else (lambda t: t.float())

    with br.fork():
        reconstructed_obs = world_model.observation_model(latent_states)
        po = {
            k: MSEDistribution(_c(reconstructed_obs[k]), dims=len(reconstructed_obs[k].shape[2:]))
            for k in cnn_keys
        }
        po.update(
            {
                k: SymlogDistribution(_c(reconstructed_obs[k]), dims=len(reconstructed_obs[k].shape[2:]))
                for k in mlp_keys
            }
        )
        observation_loss = -sum(po[k].log_prob(batch_obs[k].float()) for k in po.keys())
    with br.fork():
        pr = TwoHotEncodingDistribution(world_model.reward_model(latent_states).float(), dims=1)
        reward_loss = -pr.log_prob(rewards)
    with br.fork():
        c_logits = world_model.continue_model(latent_states).float()
        if c_logits.is_cuda and c_logits.shape == continue_targets.shape and _ops.use_hip(c_logits):
            continue_loss = continue_scale_factor * -_ops.bernoulli_log_prob(c_logits, continue_targets, 1)
        else:
            pc = td.Independent(BernoulliSafeMode(logits=c_logits), 1)
            continue_loss = continue_scale_factor * -pc.log_prob(continue_targets)
    with br.fork():
        if priors_logits is None:
            priors_logits = world_model.rssm.transition_logits(recurrent_states)
        pl = priors_logits.view(*priors_logits.shape[:-1], stochastic_size, discrete_size)
        pol = posteriors_logits.view(*posteriors_logits.shape[:-1], stochastic_size, discrete_size)
        if pol.is_cuda and _ops.use_hip(pol):
            kl_dyn_v, kl_rep_v = _ops.kl_balanced(pol.float(), pl.float())
            kl = kl_dyn_v
            free_nats = torch.full_like(kl, kl_free_nats)
            dyn_loss = kl_dynamic * torch.maximum(kl_dyn_v, free_nats)
            repr_loss = kl_representation * torch.maximum(kl_rep_v, free_nats)
        else:
            kl = dyn_loss = categorical_kl(pol.detach(), pl)
            free_nats = torch.full_like(dyn_loss, kl_free_nats)
            dyn_loss = kl_dynamic * torch.maximum(dyn_loss, free_nats)
            repr_loss = kl_representation * torch.maximum(categorical_kl(pol, pl.detach()), free_nats)
        kl_loss = dyn_loss + repr_loss
    br.join()
    rec_loss = (kl_regularizer * kl_loss + observation_loss + reward_loss + continue_loss).mean()
    return (
        rec_loss,
        kl,
        kl_loss.mean(),
        reward_loss.mean(),
        observation_loss.mean(),
        continue_loss.mean(),
        pl,
        pol,
    )
