"""Distributions.

Parity with sheeprl/utils/distribution.py:
* ``TruncatedNormal`` (:116) — DV1/DV2 continuous actor.
* ``SymlogDistribution`` (:152) — symlog-MSE log_prob for DV3 vector heads.
* ``MSEDistribution`` (:196) — DV3 image reconstruction.
* ``TwoHotEncodingDistribution`` (:224) — DV3 reward/critic: 255 bins,
  symlog transform, two-hot cross-entropy (:253-276).
* ``OneHotCategoricalST`` (:387) — straight-through one-hot categorical for
  the DV2/DV3 stochastic state.
* ``BernoulliSafeMode`` (:409) — continue head with non-NaN mode.
* ``TanhNormal`` — SAC squashed Gaussian (sac/agent.py:123-142).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributions as td
from torch import Tensor
from torch.distributions.utils import probs_to_logits

from sheeprl_amd import ops


class SymlogDistribution:
    def __init__(self, mode: Tensor, dims: int = 1, agg: str = "sum") -> None:
        self._mode = mode
        self._dims = tuple(-i for i in range(1, dims + 1))
        self._agg = agg

    @property
    def mode(self) -> Tensor:
        return ops.symexp(self._mode)

    @property
    def mean(self) -> Tensor:
        return ops.symexp(self._mode)

    def log_prob(self, value: Tensor) -> Tensor:
        if (
            self._agg == "sum"
            and self._mode.is_cuda
            and value.shape == self._mode.shape
            and not value.requires_grad
            and ops.use_hip(self._mode)
        ):
            # one reduction kernel + one elementwise backward (symlog of the
            # target folded into the kernel)
            return ops.symlog_mse_log_prob(self._mode, value, len(self._dims))
        distance = -((self._mode - ops.symlog(value)) ** 2)
        if self._agg == "mean":
            return distance.mean(self._dims)
        return distance.sum(self._dims)


class MSEDistribution:
    def __init__(self, mode: Tensor, dims: int = 1, agg: str = "sum") -> None:
        self._mode = mode
        self._dims = tuple(-i for i in range(1, dims + 1))
        self._agg = agg

    @property
    def mode(self) -> Tensor:
        return self._mode

    @property
    def mean(self) -> Tensor:
        return self._mode

    def log_prob(self, value: Tensor) -> Tensor:
        if (
            self._agg == "sum"
            and self._mode.is_cuda
            and value.shape == self._mode.shape
            and not value.requires_grad
            and ops.use_hip(self._mode)
        ):
            return ops.mse_log_prob(self._mode, value, len(self._dims))
        distance = -((self._mode - value) ** 2)
        if self._agg == "mean":
            return distance.mean(self._dims)
        return distance.sum(self._dims)


_SUPPORT_CACHE: dict = {}


def _support(low: float, high: float, k: int, device: torch.device) -> Tensor:
    key = (low, high, k, str(device))
    if key not in _SUPPORT_CACHE:
        _SUPPORT_CACHE[key] = torch.linspace(low, high, k, device=device, dtype=torch.float32)
    return _SUPPORT_CACHE[key]


class TwoHotEncodingDistribution:
    """Categorical over a symlog-spaced support; log_prob is the two-hot
    cross-entropy of the symlog'd target (reference distribution.py:224-276).
    """

    def __init__(
        self,
        logits: Tensor,
        dims: int = 1,
        low: float = -20.0,
        high: float = 20.0,
    ) -> None:
        self.logits = logits
        self._dims = tuple(-i for i in range(1, dims + 1))
        self._low, self._high = low, high
        self.bins = _support(low, high, logits.shape[-1], logits.device)

    @property
    def probs(self) -> Tensor:
        return torch.softmax(self.logits, dim=-1)

    @property
    def mean(self) -> Tensor:
        return ops.symexp((self.probs * self.bins).sum(dim=-1, keepdim=True))

    @property
    def mode(self) -> Tensor:
        return self.mean

    def log_prob(self, value: Tensor) -> Tensor:
        # value: [..., 1]; two-hot encode symlog(value) over bins
        if (
            self._dims == (-1,)
            and self.logits.is_cuda
            and self.logits.dtype == torch.float32
            and not value.requires_grad
            and ops.use_hip(self.logits)
        ):
            v = value.float()
            if v.shape[-1] == 1:
                v = v.squeeze(-1)
            return ops.twohot_log_prob(self.logits, v, self._low, self._high)
        target = ops.twohot_from_support(ops.symlog(value.float()), self.bins)
        log_pred = self.logits - torch.logsumexp(self.logits, dim=-1, keepdim=True)
        return (target * log_pred).sum(self._dims)


class OneHotCategoricalValidateArgs(td.OneHotCategorical):
    def __init__(self, probs=None, logits=None, validate_args=False):
        super().__init__(probs=probs, logits=logits, validate_args=validate_args)


def gumbel_onehot_sample(logits: Tensor) -> Tensor:
    """Gumbel-max categorical sample as a one-hot tensor.

    Same distribution as ``torch.multinomial`` sampling but built only from
    philox RNG ops (``torch.rand_like``), so it is hipGraph-capture-safe and
    avoids multinomial's sort.
    """
    u = torch.rand_like(logits)
    g = -torch.log(-torch.log(u.clamp_min(1e-20)).clamp_min(1e-20))
    idx = (logits + g).argmax(-1)
    return torch.nn.functional.one_hot(idx, logits.shape[-1]).to(logits.dtype)


class OneHotCategoricalST(td.OneHotCategorical):
    """Straight-through one-hot categorical (reference distribution.py:387)."""

    def __init__(self, probs: Optional[Tensor] = None, logits: Optional[Tensor] = None, validate_args: bool = False):
        super().__init__(probs=probs, logits=logits, validate_args=validate_args)

    def sample(self, sample_shape=torch.Size()) -> Tensor:
        if sample_shape != torch.Size():
            return super().sample(sample_shape)
        with torch.no_grad():
            return gumbel_onehot_sample(self.logits)

    def rsample(self, sample_shape=torch.Size()) -> Tensor:
        sample = self.sample(sample_shape)
        probs = self.probs
        return sample + (probs - probs.detach())


class LogProbCategorical:
    """Lightweight one-hot categorical over ALREADY-NORMALIZED log-probs
    (the fused ``categorical_st`` head's output): log_prob/entropy without
    re-normalization kernels."""

    def __init__(self, log_probs: Tensor) -> None:
        self.logits = log_probs

    def log_prob(self, onehot: Tensor) -> Tensor:
        return (self.logits * onehot.to(self.logits.dtype)).sum(-1)

    def entropy(self) -> Tensor:
        return -(self.logits.exp() * self.logits).sum(-1)

    @property
    def probs(self) -> Tensor:
        return self.logits.exp()

    @property
    def mode(self) -> Tensor:
        idx = self.logits.argmax(-1)
        return torch.nn.functional.one_hot(idx, self.logits.shape[-1]).to(self.logits.dtype)


class BernoulliSafeMode(td.Bernoulli):
    """Bernoulli whose mode is well-defined at p=0.5 (reference :409-416)."""

    def __init__(self, probs=None, logits=None, validate_args=False):
        super().__init__(probs=probs, logits=logits, validate_args=validate_args)

    @property
    def mode(self) -> Tensor:
        return (self.probs > 0.5).to(self.probs.dtype)


class TruncatedNormal(td.Distribution):
    """Normal truncated to [low, high] with reparameterized sampling via
    clamping (the Dreamer-V1/V2 actor distribution; reference :25-147 uses the
    same clamped-sample + analytic log-prob approach)."""

    arg_constraints = {}
    has_rsample = True

    def __init__(self, loc: Tensor, scale: Tensor, low: float = -1.0, high: float = 1.0, eps: float = 1e-6):
        self.loc = loc
        self.scale = scale
        self.low = low
        self.high = high
        self.eps = eps
        self._normal = td.Normal(loc, scale)
        super().__init__(self._normal.batch_shape, validate_args=False)

    def _clamp(self, x: Tensor) -> Tensor:
        clamped = x.clamp(self.low + self.eps, self.high - self.eps)
        return x - x.detach() + clamped.detach()

    def rsample(self, sample_shape=torch.Size()) -> Tensor:
        return self._clamp(self._normal.rsample(sample_shape))

    def sample(self, sample_shape=torch.Size()) -> Tensor:
        with torch.no_grad():
            return self.rsample(sample_shape)

    @property
    def mean(self) -> Tensor:
        return self._clamp(self.loc)

    @property
    def mode(self) -> Tensor:
        return self._clamp(self.loc)

    def log_prob(self, value: Tensor) -> Tensor:
        # truncation renormalization: logN(x) - log(CDF(high) - CDF(low))
        high = torch.as_tensor(self.high, dtype=value.dtype, device=value.device)
        low = torch.as_tensor(self.low, dtype=value.dtype, device=value.device)
        z = self._normal.cdf(high) - self._normal.cdf(low)
        return self._normal.log_prob(value) - torch.log(z.clamp_min(1e-8))

    def entropy(self) -> Tensor:
        # exact truncated-normal entropy (reference distribution.py:64 + log_scale):
        # H = log(sqrt(2*pi*e)*scale) + log Z - (b*phi(b) - a*phi(a)) / (2Z)
        # with a,b the standardized bounds and phi the standard-normal pdf
        a = (torch.as_tensor(self.low, dtype=self.loc.dtype, device=self.loc.device) - self.loc) / self.scale
        b = (torch.as_tensor(self.high, dtype=self.loc.dtype, device=self.loc.device) - self.loc) / self.scale
        inv_sqrt_2pi = 1.0 / math.sqrt(2.0 * math.pi)
        phi_a = torch.exp(-0.5 * a * a) * inv_sqrt_2pi
        phi_b = torch.exp(-0.5 * b * b) * inv_sqrt_2pi
        big_a = 0.5 * (1 + torch.erf(a / math.sqrt(2.0)))
        big_b = 0.5 * (1 + torch.erf(b / math.sqrt(2.0)))
        z = (big_b - big_a).clamp_min(torch.finfo(self.loc.dtype).eps)
        const = 0.5 * math.log(2.0 * math.pi * math.e)
        return const + self.scale.log() + z.log() - 0.5 * (b * phi_b - a * phi_a) / z


class TanhNormal(td.Distribution):
    """tanh-squashed Normal with the exact log-det-Jacobian correction
    (the SAC actor; reference sac/agent.py:123-142)."""

    arg_constraints = {}
    has_rsample = True

    def __init__(self, loc: Tensor, scale: Tensor):
        self.loc = loc
        self.scale = scale
        self._normal = td.Normal(loc, scale)
        super().__init__(self._normal.batch_shape, validate_args=False)

    def rsample_with_log_prob(self, sample_shape=torch.Size()):
        x = self._normal.rsample(sample_shape)
        y = torch.tanh(x)
        # log det jacobian of tanh: log(1 - tanh(x)^2) = 2*(log2 - x - softplus(-2x))
        log_prob = self._normal.log_prob(x) - 2.0 * (math.log(2.0) - x - torch.nn.functional.softplus(-2.0 * x))
        return y, log_prob

    def rsample(self, sample_shape=torch.Size()) -> Tensor:
        return torch.tanh(self._normal.rsample(sample_shape))

    def sample(self, sample_shape=torch.Size()) -> Tensor:
        with torch.no_grad():
            return self.rsample(sample_shape)

    @property
    def mode(self) -> Tensor:
        return torch.tanh(self.loc)

    def log_prob(self, value: Tensor) -> Tensor:
        x = torch.atanh(value.clamp(-1 + 1e-6, 1 - 1e-6))
        return self._normal.log_prob(x) - 2.0 * (math.log(2.0) - x - torch.nn.functional.softplus(-2.0 * x))

    def entropy(self) -> Tensor:
        return self._normal.entropy()


def unimix_logits(logits: Tensor, unimix: float = 0.01) -> Tensor:
    """1% uniform mixture on the categorical (DV3; dreamer_v3/agent.py:437-449)."""
    if unimix <= 0:
        return logits
    probs = torch.softmax(logits, dim=-1)
    uniform = torch.ones_like(probs) / probs.shape[-1]
    probs = (1 - unimix) * probs + unimix * uniform
    return probs_to_logits(probs)
