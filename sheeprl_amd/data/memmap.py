"""Disk-backed arrays for replay storage.

Parity with sheeprl/utils/memmap.py:22-270 (``MemmapArray``): a numpy memmap
with explicit file ownership, transparent pickling (the file descriptor is
dropped on ``__getstate__`` and re-opened lazily on ``__setstate__``), and
ndarray-style indexing.
"""

from __future__ import annotations

import os
import tempfile
from pathlib import Path
from typing import Any, Optional, Tuple

import numpy as np


class MemmapArray:
    def __init__(
        self,
        shape: Tuple[int, ...],
        dtype: Any = np.float32,
        mode: str = "r+",
        filename: Optional[str | Path] = None,
    ) -> None:
        self._shape = tuple(int(s) for s in shape)
        self._dtype = np.dtype(dtype)
        if filename is None:
            fd, filename = tempfile.mkstemp(suffix=".memmap")
            os.close(fd)
            self._has_ownership = True
        else:
            filename = str(filename)
            Path(filename).parent.mkdir(parents=True, exist_ok=True)
            self._has_ownership = not os.path.exists(filename)
        self._filename = str(filename)
        nbytes = int(np.prod(self._shape)) * self._dtype.itemsize
        if not os.path.exists(self._filename) or os.path.getsize(self._filename) != nbytes:
            with open(self._filename, "wb") as f:
                f.truncate(nbytes)
        self._array: Optional[np.memmap] = np.memmap(self._filename, dtype=self._dtype, mode=mode, shape=self._shape)
        self._mode = mode

    # -- factory -----------------------------------------------------------
    @classmethod
    def from_array(cls, array: np.ndarray, filename: Optional[str | Path] = None) -> "MemmapArray":
        m = cls(array.shape, array.dtype, mode="r+", filename=filename)
        m.array[:] = array
        m.array.flush()
        return m

    # -- properties ---------------------------------------------------------
    @property
    def array(self) -> np.memmap:
        if self._array is None:
            self._array = np.memmap(self._filename, dtype=self._dtype, mode="r+", shape=self._shape)
        return self._array

    @property
    def filename(self) -> str:
        return self._filename

    @property
    def has_ownership(self) -> bool:
        return self._has_ownership

    @has_ownership.setter
    def has_ownership(self, value: bool) -> None:
        self._has_ownership = bool(value)

    @property
    def shape(self) -> Tuple[int, ...]:
        return self._shape

    @property
    def dtype(self) -> np.dtype:
        return self._dtype

    def __len__(self) -> int:
        return self._shape[0]

    # -- ndarray protocol ----------------------------------------------------
    def __getitem__(self, idx: Any) -> np.ndarray:
        return self.array[idx]

    def __setitem__(self, idx: Any, value: Any) -> None:
        self.array[idx] = value

    def __array__(self, dtype: Any = None) -> np.ndarray:
        a = np.asarray(self.array)
        return a.astype(dtype) if dtype is not None else a

    # -- pickling ------------------------------------------------------------
    def __getstate__(self) -> dict:
        state = self.__dict__.copy()
        state["_array"] = None
        state["_has_ownership"] = False  # the receiving process must not delete the file
        return state

    def __setstate__(self, state: dict) -> None:
        self.__dict__.update(state)

    def __del__(self) -> None:
        if getattr(self, "_has_ownership", False) and getattr(self, "_filename", None):
            self._array = None
            try:
                os.unlink(self._filename)
            except OSError:
                pass
