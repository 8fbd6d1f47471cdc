"""Observation/action spaces (self-contained; gymnasium is not in this image).

API surface mirrors what the algorithms need from gymnasium.spaces in the
reference (Box/Discrete/MultiDiscrete/Dict): shape/dtype/sample/seed/contains.
"""

from __future__ import annotations

from typing import Any, Dict as TDict, Iterable, Optional, Tuple

import numpy as np


class Space:
    def __init__(self, shape: Optional[Tuple[int, ...]] = None, dtype: Any = None) -> None:
        self.shape = tuple(shape) if shape is not None else None
        self.dtype = np.dtype(dtype) if dtype is not None else None
        self._rng = np.random.default_rng()

    def seed(self, seed: Optional[int] = None) -> None:
        self._rng = np.random.default_rng(seed)

    def sample(self) -> Any:  # pragma: no cover - interface
        raise NotImplementedError

    def contains(self, x: Any) -> bool:  # pragma: no cover - interface
        raise NotImplementedError


class Box(Space):
    def __init__(self, low: Any, high: Any, shape: Optional[Tuple[int, ...]] = None, dtype: Any = np.float32) -> None:
        if shape is None:
            shape = np.broadcast(np.asarray(low), np.asarray(high)).shape
        super().__init__(shape, dtype)
        self.low = np.broadcast_to(np.asarray(low, dtype=self.dtype), self.shape).copy()
        self.high = np.broadcast_to(np.asarray(high, dtype=self.dtype), self.shape).copy()

    def sample(self) -> np.ndarray:
        if np.issubdtype(self.dtype, np.integer):
            return self._rng.integers(self.low, self.high.astype(np.int64) + 1, size=self.shape).astype(self.dtype)
        low = np.where(np.isfinite(self.low), self.low, -1.0)
        high = np.where(np.isfinite(self.high), self.high, 1.0)
        return (low + self._rng.random(self.shape) * (high - low)).astype(self.dtype)

    def contains(self, x: Any) -> bool:
        x = np.asarray(x)
        return x.shape == self.shape and bool(np.all(x >= self.low - 1e-6) and np.all(x <= self.high + 1e-6))

    def __repr__(self) -> str:
        return f"Box({self.shape}, {self.dtype})"


class Discrete(Space):
    def __init__(self, n: int, start: int = 0) -> None:
        super().__init__((), np.int64)
        self.n = int(n)
        self.start = int(start)

    def sample(self) -> np.int64:
        return np.int64(self.start + self._rng.integers(self.n))

    def contains(self, x: Any) -> bool:
        xi = int(x)
        return self.start <= xi < self.start + self.n

    def __repr__(self) -> str:
        return f"Discrete({self.n})"


class MultiDiscrete(Space):
    def __init__(self, nvec: Iterable[int]) -> None:
        self.nvec = np.asarray(list(nvec), dtype=np.int64)
        super().__init__(self.nvec.shape, np.int64)

    def sample(self) -> np.ndarray:
        return (self._rng.random(self.nvec.shape) * self.nvec).astype(np.int64)

    def contains(self, x: Any) -> bool:
        x = np.asarray(x)
        return x.shape == self.nvec.shape and bool(np.all(x >= 0) and np.all(x < self.nvec))

    def __repr__(self) -> str:
        return f"MultiDiscrete({self.nvec.tolist()})"


class MultiBinary(Space):
    def __init__(self, n: int) -> None:
        super().__init__((int(n),), np.int8)
        self.n = int(n)

    def sample(self) -> np.ndarray:
        return self._rng.integers(0, 2, size=(self.n,)).astype(np.int8)

    def contains(self, x: Any) -> bool:
        x = np.asarray(x)
        return x.shape == (self.n,) and bool(np.all((x == 0) | (x == 1)))


class Dict(Space):
    def __init__(self, spaces: TDict[str, Space]) -> None:
        super().__init__(None, None)
        self.spaces = dict(spaces)

    def seed(self, seed: Optional[int] = None) -> None:
        for i, s in enumerate(self.spaces.values()):
            s.seed(None if seed is None else seed + i)

    def sample(self) -> TDict[str, Any]:
        return {k: s.sample() for k, s in self.spaces.items()}

    def contains(self, x: Any) -> bool:
        return isinstance(x, dict) and all(k in x and s.contains(x[k]) for k, s in self.spaces.items())

    def keys(self):
        return self.spaces.keys()

    def items(self):
        return self.spaces.items()

    def values(self):
        return self.spaces.values()

    def __getitem__(self, k: str) -> Space:
        return self.spaces[k]

    def __contains__(self, k: str) -> bool:
        return k in self.spaces

    def __repr__(self) -> str:
        return f"Dict({self.spaces})"
