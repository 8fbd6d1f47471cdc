import math

import pytest
import torch

from sheeprl_amd import ops
from sheeprl_amd.distributions import (
    BernoulliSafeMode,
    LogProbCategorical,
    MSEDistribution,
    OneHotCategoricalST,
    SymlogDistribution,
    TanhNormal,
    TruncatedNormal,
    TwoHotEncodingDistribution,
    unimix_logits,
)


def test_twohot_distribution_mean_roundtrip():
    # put all mass on one bin: mean must symexp the bin value
    logits = torch.full((3, 255), -1e9)
    logits[:, 127] = 0.0  # center bin = 0 -> symexp(0)=0
    d = TwoHotEncodingDistribution(logits, dims=1)
    assert torch.allclose(d.mean, torch.zeros(3, 1), atol=1e-4)


def test_twohot_distribution_log_prob_is_ce():
    torch.manual_seed(0)
    logits = torch.randn(5, 255)
    x = torch.randn(5, 1) * 3
    d = TwoHotEncodingDistribution(logits, dims=1)
    target = ops.twohot_from_support(ops.symlog(x), d.bins)
    expected = (target * torch.log_softmax(logits, -1)).sum(-1)
    assert torch.allclose(d.log_prob(x), expected, atol=1e-5)


def test_onehot_st_sample_is_onehot_and_st():
    torch.manual_seed(1)
    logits = torch.randn(64, 8, requires_grad=True)
    d = OneHotCategoricalST(logits=logits)
    s = d.rsample()
    assert torch.all(s.sum(-1) == 1)
    s.sum().backward()
    assert logits.grad is not None  # straight-through path exists


def test_bernoulli_safe_mode_at_half():
    d = BernoulliSafeMode(probs=torch.full((4,), 0.5))
    assert torch.all(torch.isfinite(d.mode))


def test_tanh_normal_log_prob_matches_transformed():
    torch.manual_seed(2)
    loc = torch.randn(10)
    scale = torch.rand(10) + 0.1
    d = TanhNormal(loc, scale)
    y, logp = d.rsample_with_log_prob()
    ref = torch.distributions.TransformedDistribution(
        torch.distributions.Normal(loc, scale), torch.distributions.TanhTransform()
    )
    assert torch.allclose(logp, ref.log_prob(y.clamp(-1 + 1e-6, 1 - 1e-6)), atol=1e-4)


def test_truncated_normal_samples_in_bounds():
    d = TruncatedNormal(torch.zeros(100), torch.ones(100) * 3)
    s = d.rsample()
    assert s.min() >= -1 and s.max() <= 1


def test_unimix_logits_mixture():
    logits = torch.tensor([[100.0, 0.0]])
    mixed = unimix_logits(logits, unimix=0.01)
    p = torch.softmax(mixed, -1)
    assert p[0, 1] >= 0.005 - 1e-6  # uniform floor


def test_logprob_categorical_consistency():
    torch.manual_seed(3)
    raw = torch.randn(6, 5)
    m, onehot = ops.categorical_st(raw, unimix=0.01, sample=False)
    d = LogProbCategorical(m)
    ref = torch.distributions.OneHotCategorical(logits=m)
    assert torch.allclose(d.log_prob(onehot.float()), ref.log_prob(onehot.float()), atol=1e-5)
    assert torch.allclose(d.entropy(), ref.entropy(), atol=1e-5)


def test_symlog_mse_distributions():
    x = torch.randn(4, 7)
    d = SymlogDistribution(ops.symlog(x), dims=1)
    assert torch.allclose(d.log_prob(x), torch.zeros(4), atol=1e-5)  # perfect prediction
    m = MSEDistribution(x, dims=1)
    assert torch.allclose(m.log_prob(x), torch.zeros(4), atol=1e-6)


def test_truncated_normal_entropy_numeric():
    # exact truncated entropy vs numeric integration of -p log p over [low, high]
    import numpy as np
    from scipy import stats

    loc = torch.tensor([0.0, 0.5, -0.8])
    scale = torch.tensor([1.0, 0.3, 2.5])
    d = TruncatedNormal(loc, scale)
    ent = d.entropy()
    for i in range(3):
        a = (-1.0 - loc[i].item()) / scale[i].item()
        b = (1.0 - loc[i].item()) / scale[i].item()
        ref = stats.truncnorm.entropy(a, b, loc=loc[i].item(), scale=scale[i].item())
        assert abs(ent[i].item() - float(ref)) < 1e-4, (i, ent[i].item(), ref)
