"""Replay buffers.

Behavioral parity with sheeprl/data/buffers.py:

* :class:`ReplayBuffer` (buffers.py:20-360) — circular dict-of-arrays
  ``[buffer_size, n_envs, ...]``, wrap-around ``add``, uniform ``sample`` with
  ``sample_next_obs`` index arithmetic that avoids the invalid write head.
* :class:`SequentialReplayBuffer` (buffers.py:363) — contiguous
  ``sequence_length`` windows ignoring episode bounds; output
  ``[n_samples, sequence_length, batch_size, ...]``.
* :class:`EnvIndependentReplayBuffer` (buffers.py:529) — one sub-buffer per
  env with independent wrap positions; sampling splits the batch across
  sub-buffers with ``np.bincount``.
* :class:`EpisodeBuffer` (buffers.py:746) — whole-episode storage with
  per-env open episodes, terminated/truncated splitting, oldest-episode
  eviction and ``prioritize_ends`` sampling.

MI355X note: host storage is plain numpy (optionally memmap); the GPU path
converts samples through :func:`sample_tensors` into pinned staging buffers
with async H2D copies on a side stream (see sheeprl_amd/data/prefetch.py).
"""

from __future__ import annotations

from pathlib import Path
from typing import Any, Dict, List, Optional, Sequence

import numpy as np
import torch

from sheeprl_amd.data.memmap import MemmapArray
from sheeprl_amd.utils.utils import NUMPY_TO_TORCH_DTYPE_DICT


def get_tensor(
    array: np.ndarray,
    dtype: Optional[torch.dtype] = None,
    clone: bool = False,
    device: str | torch.device = "cpu",
    from_numpy: bool = False,
) -> torch.Tensor:
    """numpy -> torch with the reference's dtype map (buffers.py:1158-1180)."""
    arr = np.asarray(array)
    torch_dtype = dtype or NUMPY_TO_TORCH_DTYPE_DICT.get(arr.dtype, torch.float32)
    if from_numpy:
        t = torch.from_numpy(np.ascontiguousarray(arr))
    else:
        t = torch.as_tensor(np.ascontiguousarray(arr))
    t = t.to(device=device, dtype=torch_dtype)
    if clone and not from_numpy:
        t = t.clone()
    return t


class ReplayBuffer:
    batch_axis: int = 1  # concat axis used by EnvIndependentReplayBuffer

    def __init__(
        self,
        buffer_size: int,
        n_envs: int = 1,
        obs_keys: Sequence[str] = ("observations",),
        memmap: bool = False,
        memmap_dir: Optional[str | Path] = None,
        pinned: bool = False,
        **kwargs: Any,
    ) -> None:
        if buffer_size <= 0:
            raise ValueError(f"buffer_size must be > 0, got {buffer_size}")
        if n_envs <= 0:
            raise ValueError(f"n_envs must be > 0, got {n_envs}")
        self._buffer_size = buffer_size
        self._n_envs = n_envs
        self._obs_keys = tuple(obs_keys)
        self._memmap = memmap
        # pinned host storage: the ring becomes device-readable zero-copy so
        # the HIP replay-gather kernel can pull sequence windows straight into
        # HBM; the numpy view shares the pinned
        # torch storage, so every host-side add/sample path is unchanged
        self._pinned = bool(pinned) and not memmap and torch.cuda.is_available()
        self._pinned_t: Dict[str, torch.Tensor] = {}
        self._memmap_dir = Path(memmap_dir) if memmap_dir is not None else None
        if memmap and self._memmap_dir is not None:
            self._memmap_dir.mkdir(parents=True, exist_ok=True)
        self._buf: Dict[str, Any] = {}
        self._pos = 0
        self._full = False

    # -- properties ---------------------------------------------------------
    @property
    def buffer(self) -> Dict[str, np.ndarray]:
        return {k: np.asarray(v) for k, v in self._buf.items()}

    @property
    def buffer_size(self) -> int:
        return self._buffer_size

    @property
    def n_envs(self) -> int:
        return self._n_envs

    @property
    def full(self) -> bool:
        return self._full

    @property
    def is_memmap(self) -> bool:
        return self._memmap

    @property
    def empty(self) -> bool:
        return len(self) == 0

    def __len__(self) -> int:
        return self._buffer_size if self._full else self._pos

    def __contains__(self, key: str) -> bool:
        return key in self._buf

    def __getitem__(self, key: str) -> np.ndarray:
        return np.asarray(self._buf[key])

    def __setitem__(self, key: str, value: np.ndarray) -> None:
        self._alloc(key, value.shape[2:], value.dtype)
        self._buf[key][:] = value

    # -- storage ------------------------------------------------------------
    def _alloc(self, key: str, item_shape: tuple, dtype: Any) -> None:
        if key in self._buf:
            return
        shape = (self._buffer_size, self._n_envs, *item_shape)
        if self._memmap:
            fn = None if self._memmap_dir is None else self._memmap_dir / f"{key}.memmap"
            self._buf[key] = MemmapArray(shape, dtype, filename=fn)
        elif self._pinned:
            tdtype = {"uint8": torch.uint8, "float32": torch.float32, "float64": torch.float64,
                      "int64": torch.int64, "int32": torch.int32, "bool": torch.bool}.get(
                np.dtype(dtype).name)
            if tdtype is None:
                self._buf[key] = np.zeros(shape, dtype=dtype)
            else:
                t = torch.zeros(shape, dtype=tdtype, pin_memory=True)
                self._pinned_t[key] = t
                self._buf[key] = t.numpy()
        else:
            self._buf[key] = np.zeros(shape, dtype=dtype)

    def add(self, data: Dict[str, np.ndarray], validate_args: bool = True) -> None:
        """``data[k]`` has shape ``[seq_len, n_envs, ...]``; written circularly."""
        if validate_args:
            if not isinstance(data, dict) or not data:
                raise ValueError("add expects a non-empty dict of arrays")
            lens = {k: v.shape[:2] for k, v in data.items()}
            first = next(iter(lens.values()))
            if any(v != first for v in lens.values()):
                raise RuntimeError(f"all arrays must share [seq_len, n_envs]: {lens}")
            if first[1] != self._n_envs:
                raise RuntimeError(f"expected n_envs={self._n_envs}, got {first[1]}")
        seq_len = next(iter(data.values())).shape[0]
        for k, v in data.items():
            self._alloc(k, v.shape[2:], v.dtype)
        if seq_len > self._buffer_size:
            data = {k: v[-self._buffer_size :] for k, v in data.items()}
            seq_len = self._buffer_size
        idxes = (self._pos + np.arange(seq_len)) % self._buffer_size
        for k, v in data.items():
            self._buf[k][idxes] = v
        self._full = self._full or (self._pos + seq_len >= self._buffer_size)
        self._pos = int((self._pos + seq_len) % self._buffer_size)

    # -- sampling -----------------------------------------------------------
    def _valid_len(self) -> int:
        return self._buffer_size if self._full else self._pos

    def sample(
        self,
        batch_size: int,
        sample_next_obs: bool = False,
        clone: bool = False,
        n_samples: int = 1,
        **kwargs: Any,
    ) -> Dict[str, np.ndarray]:
        """Uniform sample; returns arrays ``[n_samples, batch_size, ...]``."""
        if batch_size <= 0 or n_samples <= 0:
            raise ValueError("batch_size and n_samples must be > 0")
        if not self._full and self._pos == 0:
            raise ValueError("cannot sample from an empty buffer")
        total = batch_size * n_samples
        valid = self._valid_len()
        if self._full and sample_next_obs:
            # the transition written at _pos-1 has its next obs at _pos, which
            # is the (stale) oldest slot: exclude it (reference buffers.py:248-264)
            offs = np.random.randint(0, self._buffer_size - 1, size=total)
            idxes = (self._pos + offs) % self._buffer_size
        elif self._full:
            idxes = np.random.randint(0, self._buffer_size, size=total)
        else:
            hi = valid - 1 if sample_next_obs else valid
            if hi <= 0:
                raise ValueError("not enough data to sample next observations")
            idxes = np.random.randint(0, hi, size=total)
        env_idxes = np.random.randint(0, self._n_envs, size=total)
        out: Dict[str, np.ndarray] = {}
        for k, v in self._buf.items():
            arr = np.asarray(v)[idxes, env_idxes]
            out[k] = arr.reshape(n_samples, batch_size, *arr.shape[1:])
            if clone:
                out[k] = out[k].copy()
        if sample_next_obs:
            nxt = (idxes + 1) % self._buffer_size
            for k in self._obs_keys:
                if k in self._buf:
                    arr = np.asarray(self._buf[k])[nxt, env_idxes]
                    out[f"next_{k}"] = arr.reshape(n_samples, batch_size, *arr.shape[1:])
        return out

    def sample_tensors(
        self,
        batch_size: int,
        sample_next_obs: bool = False,
        n_samples: int = 1,
        dtype: Optional[torch.dtype] = None,
        device: str | torch.device = "cpu",
        from_numpy: bool = False,
        **kwargs: Any,
    ) -> Dict[str, torch.Tensor]:
        samples = self.sample(batch_size, sample_next_obs=sample_next_obs, n_samples=n_samples, **kwargs)
        return {k: get_tensor(v, dtype=None, device=device, from_numpy=from_numpy) for k, v in samples.items()}

    def to_tensor(
        self, dtype: Optional[torch.dtype] = None, device: str | torch.device = "cpu", from_numpy: bool = False
    ) -> Dict[str, torch.Tensor]:
        return {k: get_tensor(np.asarray(v), dtype, device=device, from_numpy=from_numpy) for k, v in self._buf.items()}

    # -- checkpoint ----------------------------------------------------------
    def state_dict(self) -> Dict[str, Any]:
        return {
            "buffer": {k: np.asarray(v).copy() for k, v in self._buf.items()},
            "pos": self._pos,
            "full": self._full,
        }

    def load_state_dict(self, state: Dict[str, Any]) -> "ReplayBuffer":
        for k, v in state["buffer"].items():
            self._alloc(k, v.shape[2:], v.dtype)
            self._buf[k][:] = v
        self._pos = state["pos"]
        self._full = state["full"]
        return self


class SequentialReplayBuffer(ReplayBuffer):
    """Samples contiguous time windows (reference buffers.py:363-526).

    Output arrays are ``[n_samples, sequence_length, batch_size, ...]``.
    Windows never cross the write head when the buffer is full.
    """

    batch_axis: int = 2

    def sample(  # type: ignore[override]
        self,
        batch_size: int,
        sample_next_obs: bool = False,
        clone: bool = False,
        n_samples: int = 1,
        sequence_length: int = 1,
        **kwargs: Any,
    ) -> Dict[str, np.ndarray]:
        if batch_size <= 0 or n_samples <= 0:
            raise ValueError("batch_size and n_samples must be > 0")
        L = sequence_length
        valid = self._valid_len()
        if valid < L:
            raise ValueError(f"too few samples ({valid}) for sequence_length={L}")
        total = batch_size * n_samples
        if self._full:
            # valid starts: any slot such that the window [s, s+L) does not
            # cross the write head _pos (reference buffers.py:439-460)
            n_starts = self._buffer_size - L + 1
            offs = np.random.randint(0, n_starts, size=total)
            starts = (self._pos + offs) % self._buffer_size
        else:
            starts = np.random.randint(0, valid - L + 1, size=total)
        env_idxes = np.random.randint(0, self._n_envs, size=total)
        self._last_picks = (starts.copy(), env_idxes.copy())  # device-gather hook
        win = (starts[:, None] + np.arange(L)[None, :]) % self._buffer_size  # [total, L]
        out: Dict[str, np.ndarray] = {}
        for k, v in self._buf.items():
            arr = np.asarray(v)[win, env_idxes[:, None]]  # [total, L, ...]
            arr = arr.reshape(n_samples, batch_size, L, *arr.shape[2:]).swapaxes(1, 2)
            out[k] = arr.copy() if clone else arr
        return out


class EnvIndependentReplayBuffer:
    """One sub-buffer per env, independent wrap positions
    (reference buffers.py:529-743)."""

    def __init__(
        self,
        buffer_size: int,
        n_envs: int = 1,
        obs_keys: Sequence[str] = ("observations",),
        memmap: bool = False,
        memmap_dir: Optional[str | Path] = None,
        buffer_cls: type = SequentialReplayBuffer,
        **kwargs: Any,
    ) -> None:
        self._buffer_size = buffer_size
        self._n_envs = n_envs
        self._buffer_cls = buffer_cls
        self._concat_along_axis = buffer_cls.batch_axis
        self._buf: List[ReplayBuffer] = [
            buffer_cls(
                buffer_size,
                n_envs=1,
                obs_keys=obs_keys,
                memmap=memmap,
                memmap_dir=None if memmap_dir is None else Path(memmap_dir) / f"env_{i}",
                **kwargs,
            )
            for i in range(n_envs)
        ]

    @property
    def buffer(self) -> List[ReplayBuffer]:
        return self._buf

    @property
    def n_envs(self) -> int:
        return self._n_envs

    @property
    def buffer_size(self) -> int:
        return self._buffer_size

    @property
    def full(self) -> bool:
        return all(b.full for b in self._buf)

    def __len__(self) -> int:
        return sum(len(b) for b in self._buf)

    def add(self, data: Dict[str, np.ndarray], indices: Optional[Sequence[int]] = None) -> None:
        if indices is None:
            indices = range(self._n_envs)
        for pos, env_idx in enumerate(indices):
            self._buf[env_idx].add({k: v[:, pos : pos + 1] for k, v in data.items()})

    def sample(self, batch_size: int, n_samples: int = 1, **kwargs: Any) -> Dict[str, np.ndarray]:
        if batch_size <= 0 or n_samples <= 0:
            raise ValueError("batch_size and n_samples must be > 0")
        nonempty = [i for i, b in enumerate(self._buf) if len(b) > 0]
        L = kwargs.get("sequence_length", 1)
        nonempty = [i for i in nonempty if len(self._buf[i]) >= L]
        if not nonempty:
            raise ValueError("no sub-buffer has enough data to sample")
        picks = np.random.choice(nonempty, size=batch_size)
        counts = np.bincount(picks, minlength=self._n_envs)
        parts: List[Dict[str, np.ndarray]] = []
        for i, c in enumerate(counts):
            if c == 0:
                continue
            parts.append(self._buf[i].sample(int(c), n_samples=n_samples, **kwargs))
        keys = parts[0].keys()
        return {k: np.concatenate([p[k] for p in parts], axis=self._concat_along_axis) for k in keys}

    def sample_tensors(
        self,
        batch_size: int,
        n_samples: int = 1,
        dtype: Optional[torch.dtype] = None,
        device: str | torch.device = "cpu",
        from_numpy: bool = False,
        **kwargs: Any,
    ) -> Dict[str, torch.Tensor]:
        s = self.sample(batch_size, n_samples=n_samples, **kwargs)
        return {k: get_tensor(v, device=device, from_numpy=from_numpy) for k, v in s.items()}

    def state_dict(self) -> Dict[str, Any]:
        return {"buffers": [b.state_dict() for b in self._buf]}

    def load_state_dict(self, state: Dict[str, Any]) -> "EnvIndependentReplayBuffer":
        for b, s in zip(self._buf, state["buffers"]):
            b.load_state_dict(s)
        return self


class EpisodeBuffer:
    """Whole-episode storage (reference buffers.py:746-1155).

    ``add`` receives chunks ``[seq_len, n_envs, ...]`` and splits them into
    episodes on the ``terminated``/``truncated`` flags; complete episodes of
    length >= ``minimum_episode_length`` are committed, the oldest episodes are
    evicted when total stored steps exceed ``buffer_size``.  ``sample`` draws
    fixed-length windows inside episodes, optionally prioritizing episode ends.
    """

    def __init__(
        self,
        buffer_size: int,
        sequence_length: int,
        n_envs: int = 1,
        obs_keys: Sequence[str] = ("observations",),
        prioritize_ends: bool = False,
        memmap: bool = False,
        memmap_dir: Optional[str | Path] = None,
        minimum_episode_length: Optional[int] = None,
        **kwargs: Any,
    ) -> None:
        if buffer_size <= 0:
            raise ValueError(f"buffer_size must be > 0, got {buffer_size}")
        if sequence_length <= 0:
            raise ValueError(f"sequence_length must be > 0, got {sequence_length}")
        if buffer_size < sequence_length:
            raise ValueError(f"buffer_size ({buffer_size}) must be >= sequence_length ({sequence_length})")
        self._buffer_size = buffer_size
        self._sequence_length = sequence_length
        self._minimum_episode_length = minimum_episode_length or sequence_length
        self._n_envs = n_envs
        self._obs_keys = tuple(obs_keys)
        self._prioritize_ends = prioritize_ends
        self._memmap = memmap
        self._pinned = bool(kwargs.get("pinned", False)) and not memmap and torch.cuda.is_available()
        self._pinned_t: Dict[str, torch.Tensor] = {}
        self._memmap_dir = Path(memmap_dir) if memmap_dir is not None else None
        if memmap and self._memmap_dir is not None:
            self._memmap_dir.mkdir(parents=True, exist_ok=True)
        self._episode_counter = 0
        self._episodes: List[Dict[str, np.ndarray]] = []
        self._open: List[Optional[Dict[str, List[np.ndarray]]]] = [None] * n_envs
        self._cum_lengths: List[int] = []

    @property
    def is_memmap(self) -> bool:
        return self._memmap

    @property
    def buffer(self) -> List[Dict[str, np.ndarray]]:
        return self._episodes

    @property
    def n_envs(self) -> int:
        return self._n_envs

    @property
    def buffer_size(self) -> int:
        return self._buffer_size

    @property
    def sequence_length(self) -> int:
        return self._sequence_length

    @property
    def full(self) -> bool:
        return len(self) >= self._buffer_size

    def __len__(self) -> int:
        return sum(ep[next(iter(ep))].shape[0] for ep in self._episodes)

    def add(self, data: Dict[str, np.ndarray], indices: Optional[Sequence[int]] = None) -> None:
        if "terminated" not in data or "truncated" not in data:
            raise RuntimeError("EpisodeBuffer.add requires 'terminated' and 'truncated' keys")
        if indices is None:
            indices = range(self._n_envs)
        seq_len = next(iter(data.values())).shape[0]
        done = np.logical_or(data["terminated"], data["truncated"])  # [seq, n]
        for pos, env_idx in enumerate(indices):
            if self._open[env_idx] is None:
                self._open[env_idx] = {k: [] for k in data}
            open_ep = self._open[env_idx]
            start = 0
            env_done = done[:, pos].reshape(-1)
            ends = list(np.nonzero(env_done)[0])
            for end in ends:
                for k, v in data.items():
                    open_ep[k].append(v[start : end + 1, pos])
                self._commit(env_idx)
                self._open[env_idx] = {k: [] for k in data}
                open_ep = self._open[env_idx]
                start = end + 1
            if start < seq_len:
                for k, v in data.items():
                    open_ep[k].append(v[start:, pos])

    def _commit(self, env_idx: int) -> None:
        open_ep = self._open[env_idx]
        if open_ep is None:
            return
        ep = {k: np.concatenate(v, axis=0) if v else np.empty((0,)) for k, v in open_ep.items()}
        length = ep[next(iter(ep))].shape[0]
        if length < self._minimum_episode_length:
            return
        if length > self._buffer_size:
            ep = {k: v[-self._buffer_size :] for k, v in ep.items()}
            length = self._buffer_size
        if self._memmap and self._memmap_dir is not None:
            # spill the committed episode to disk (reference buffers.py:961-991):
            # one MemmapArray file per key per episode; only the sampled
            # windows are faulted back in
            ep_dir = self._memmap_dir / f"episode_{self._episode_counter}"
            ep_dir.mkdir(parents=True, exist_ok=True)
            self._episode_counter += 1
            spilled = {}
            for k, v in ep.items():
                ma = MemmapArray.from_array(v, filename=ep_dir / f"{k}.memmap")
                ma.has_ownership = False  # eviction unlinks explicitly
                spilled[k] = ma.array
            ep = spilled
        self._episodes.append(ep)
        # evict the oldest episodes (incl. memmap files, reference
        # buffers.py:993-1014) until total length fits
        while len(self) > self._buffer_size and len(self._episodes) > 1:
            old_ep = self._episodes.pop(0)
            self._evict_files(old_ep)
        if len(self) > self._buffer_size:
            ep = self._episodes[0]
            self._episodes[0] = {k: np.asarray(v)[-self._buffer_size :] for k, v in ep.items()}

    @staticmethod
    def _evict_files(ep: Dict[str, np.ndarray]) -> None:
        import os

        for v in ep.values():
            fn = getattr(v, "filename", None)
            if fn:
                try:
                    os.unlink(fn)
                except OSError:
                    pass

    def sample(
        self,
        batch_size: int,
        n_samples: int = 1,
        clone: bool = False,
        sequence_length: Optional[int] = None,
        prioritize_ends: Optional[bool] = None,
        **kwargs: Any,
    ) -> Dict[str, np.ndarray]:
        if batch_size <= 0 or n_samples <= 0:
            raise ValueError("batch_size and n_samples must be > 0")
        L = sequence_length or self._sequence_length
        pe = self._prioritize_ends if prioritize_ends is None else prioritize_ends
        lengths = np.array([ep[next(iter(ep))].shape[0] for ep in self._episodes])
        ok = np.nonzero(lengths >= L)[0]
        if len(ok) == 0:
            raise RuntimeError(f"no episodes of length >= {L} to sample")
        w = lengths[ok].astype(np.float64)
        w /= w.sum()
        total = batch_size * n_samples
        picks = np.random.choice(ok, size=total, p=w)
        out_parts: Dict[str, List[np.ndarray]] = {}
        for idx in picks:
            ep = self._episodes[idx]
            elen = lengths[idx]
            if pe:
                start = np.random.randint(0, elen)
                start = min(start, elen - L)
            else:
                start = np.random.randint(0, elen - L + 1)
            for k, v in ep.items():
                out_parts.setdefault(k, []).append(v[start : start + L])
        out: Dict[str, np.ndarray] = {}
        for k, lst in out_parts.items():
            arr = np.stack(lst)  # [total, L, ...]
            arr = arr.reshape(n_samples, batch_size, L, *arr.shape[2:]).swapaxes(1, 2)
            out[k] = arr.copy() if clone else arr
        return out

    def sample_tensors(
        self,
        batch_size: int,
        n_samples: int = 1,
        device: str | torch.device = "cpu",
        from_numpy: bool = False,
        **kwargs: Any,
    ) -> Dict[str, torch.Tensor]:
        s = self.sample(batch_size, n_samples=n_samples, **kwargs)
        return {k: get_tensor(v, device=device, from_numpy=from_numpy) for k, v in s.items()}

    def state_dict(self) -> Dict[str, Any]:
        return {"episodes": self._episodes, "open": self._open}

    def load_state_dict(self, state: Dict[str, Any]) -> "EpisodeBuffer":
        self._episodes = state["episodes"]
        self._open = state.get("open", [None] * self._n_envs)
        return self
