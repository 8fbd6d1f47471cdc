"""Metric aggregation without the torchmetrics dependency.

Parity surface with the reference (sheeprl/utils/metric.py):
* :class:`MeanMetric` / :class:`SumMetric` / :class:`MaxMetric` — the metric
  primitives the reference takes from torchmetrics (not in this image).
* :class:`MetricAggregator` (metric.py:17-143) — named-metric dict with
  update/compute/reset, NaN dropping, global disable.
* :class:`RankIndependentMetricAggregator` (metric.py:146-195) — no implicit
  cross-rank sync; ``compute`` all-gathers the computed values.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import torch


class Metric:
    sync_on_compute: bool = False

    def update(self, value: Any) -> None:  # pragma: no cover - interface
        raise NotImplementedError

    def compute(self) -> Any:  # pragma: no cover - interface
        raise NotImplementedError

    def reset(self) -> None:  # pragma: no cover - interface
        raise NotImplementedError


def _to_float(value: Any) -> float:
    if isinstance(value, torch.Tensor):
        return float(value.detach().float().mean().item() if value.numel() > 1 else value.detach().float().item())
    return float(value)


class MeanMetric(Metric):
    def __init__(self, sync_on_compute: bool = False) -> None:
        self.sync_on_compute = sync_on_compute
        self._sum = 0.0
        self._count = 0

    def update(self, value: Any) -> None:
        v = _to_float(value)
        if v == v:  # drop NaN
            self._sum += v
            self._count += 1

    def compute(self) -> float:
        if self._count == 0:
            return float("nan")
        s, c = self._sum, self._count
        if self.sync_on_compute:
            s, c = _sync_sum(s), _sync_sum(float(c))
        return s / max(c, 1)

    def reset(self) -> None:
        self._sum = 0.0
        self._count = 0


class SumMetric(Metric):
    def __init__(self, sync_on_compute: bool = False) -> None:
        self.sync_on_compute = sync_on_compute
        self._sum = 0.0

    def update(self, value: Any) -> None:
        v = _to_float(value)
        if v == v:
            self._sum += v

    def compute(self) -> float:
        return _sync_sum(self._sum) if self.sync_on_compute else self._sum

    def reset(self) -> None:
        self._sum = 0.0


class MaxMetric(Metric):
    def __init__(self, sync_on_compute: bool = False) -> None:
        self.sync_on_compute = sync_on_compute
        self._max: Optional[float] = None

    def update(self, value: Any) -> None:
        v = _to_float(value)
        if v == v:
            self._max = v if self._max is None else max(self._max, v)

    def compute(self) -> float:
        return float("nan") if self._max is None else self._max

    def reset(self) -> None:
        self._max = None


def _sync_sum(v: float) -> float:
    import torch.distributed as dist

    if dist.is_available() and dist.is_initialized():
        t = torch.tensor([v], dtype=torch.float64)
        dist.all_reduce(t)
        return float(t.item())
    return v


_METRIC_TYPES = {"mean": MeanMetric, "sum": SumMetric, "max": MaxMetric}


def make_metric(spec: Any, sync_on_compute: bool = False) -> Metric:
    if isinstance(spec, Metric):
        return spec
    if isinstance(spec, str):
        return _METRIC_TYPES[spec](sync_on_compute=sync_on_compute)
    if isinstance(spec, dict):
        kind = spec.get("kind", "mean")
        return _METRIC_TYPES[kind](sync_on_compute=spec.get("sync_on_compute", sync_on_compute))
    raise TypeError(f"cannot build metric from {spec!r}")


class MetricAggregator:
    disabled: bool = False

    def __init__(self, metrics: Optional[Dict[str, Any]] = None, sync_on_compute: bool = False) -> None:
        self.sync_on_compute = sync_on_compute
        self.metrics: Dict[str, Metric] = {}
        for name, spec in (metrics or {}).items():
            self.metrics[name] = make_metric(spec, sync_on_compute)

    def add(self, name: str, spec: Any = "mean") -> None:
        if name not in self.metrics:
            self.metrics[name] = make_metric(spec, self.sync_on_compute)

    def update(self, name: str, value: Any) -> None:
        if MetricAggregator.disabled:
            return
        if name not in self.metrics:
            self.add(name)
        self.metrics[name].update(value)

    def compute(self) -> Dict[str, float]:
        if MetricAggregator.disabled:
            return {}
        out: Dict[str, float] = {}
        for name, m in self.metrics.items():
            v = m.compute()
            if v == v:  # drop NaN (empty metrics)
                out[name] = v
        return out

    def reset(self) -> None:
        for m in self.metrics.values():
            m.reset()

    def to(self, device: Any) -> "MetricAggregator":
        return self


class RankIndependentMetricAggregator(MetricAggregator):
    """Aggregator whose ``compute`` returns the all-gathered per-rank values."""

    def __init__(self, metrics: Optional[Dict[str, Any]] = None) -> None:
        super().__init__(metrics, sync_on_compute=False)

    def compute(self) -> List[Dict[str, float]]:  # type: ignore[override]
        import torch.distributed as dist

        local = super().compute()
        if dist.is_available() and dist.is_initialized():
            out: List[Optional[Dict[str, float]]] = [None] * dist.get_world_size()
            dist.all_gather_object(out, local)
            return [x for x in out if x is not None]
        return [local]
