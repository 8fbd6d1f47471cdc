"""Device-side replay gather (BASELINE north star).

The replay ring lives in PINNED host memory (``ReplayBuffer(pinned=True)``);
a HIP kernel (``ops ext replay_gather``) reads the rings zero-copy over PCIe
and writes sequence windows straight into HBM on a SIDE stream, double-
buffered one batch ahead of the (graph-replayed) gradient step.  This
replaces the retired Python prefetch thread (round-1: GIL + memcpy
contention measured ~10% slower) and the numpy fancy-index + H2D path
(reference buffers.py:493-511).
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np
import torch

from sheeprl_amd.data.buffers import EnvIndependentReplayBuffer
from sheeprl_amd.ops._ext import require_ext
from sheeprl_amd.utils.utils import NUMPY_TO_TORCH_DTYPE_DICT


def _base_ptr(arr: np.ndarray) -> int:
    return int(arr.__array_interface__["data"][0])


class DeviceReplayGather:
    """Double-buffered HIP gather of ``[1, L, B, ...]`` sequence batches from
    an :class:`EnvIndependentReplayBuffer` whose sub-buffers are pinned."""

    def __init__(self, rb: EnvIndependentReplayBuffer, batch_size: int, seq_len: int,
                 device: torch.device, depth: int = 2) -> None:
        self._rb = rb
        self._bs = batch_size
        self._L = seq_len
        self._dev = device
        self._ext = require_ext()
        self._stream = torch.cuda.Stream()
        self._depth = depth
        self._outs: List[Dict[str, torch.Tensor]] = []
        self._events = [torch.cuda.Event() for _ in range(depth)]
        self._cur = 0
        self._primed = False
        sub0 = rb._buf[0]
        if not getattr(sub0, "_pinned", False):
            raise RuntimeError("DeviceReplayGather needs ReplayBuffer(pinned=True) sub-buffers")
        self._keys = list(sub0._buf.keys()) if sub0._buf else None
        self._cap = rb._buffer_size

    def _ensure_outs(self) -> None:
        if self._outs:
            return
        sub0 = self._rb._buf[0]
        self._keys = list(sub0._buf.keys())
        for _ in range(self._depth):
            out = {}
            for k in self._keys:
                arr = np.asarray(sub0._buf[k])
                item = arr.shape[2:]
                out[k] = torch.empty(
                    1, self._L, self._bs, *item,
                    dtype=NUMPY_TO_TORCH_DTYPE_DICT[np.dtype(arr.dtype)],
                    device=self._dev,
                )
            self._outs.append(out)

    def _pick(self):
        """(env, start) pairs replicating SequentialReplayBuffer's valid-window
        rule per sub-buffer (reference buffers.py:439-460)."""
        L = self._L
        nonempty = [i for i, b in enumerate(self._rb._buf) if len(b) >= L]
        if not nonempty:
            raise ValueError("no sub-buffer has enough data for the device gather")
        envs = np.random.choice(nonempty, size=self._bs)
        starts = np.empty(self._bs, dtype=np.int64)
        for j, e in enumerate(envs):
            b = self._rb._buf[int(e)]
            if b._full:
                starts[j] = (b._pos + np.random.randint(0, b._buffer_size - L + 1)) % b._buffer_size
            else:
                starts[j] = np.random.randint(0, len(b) - L + 1)
        return envs, starts

    def _launch(self, slot: int) -> None:
        self._ensure_outs()
        envs, starts = self._pick()
        starts_t = torch.from_numpy(starts).pin_memory()
        with torch.cuda.stream(self._stream):
            starts_d = starts_t.to(self._dev, non_blocking=True)
            for k in self._keys:
                ptrs = np.array(
                    [_base_ptr(np.asarray(self._rb._buf[int(e)]._buf[k])) for e in envs], dtype=np.int64
                )
                ptrs_d = torch.from_numpy(ptrs).pin_memory().to(self._dev, non_blocking=True)
                out = self._outs[slot][k]
                row_bytes = out[0, 0, 0].numel() * out.element_size()
                self._ext.replay_gather(
                    ptrs_d, starts_d, out.view(-1), self._L, self._cap, row_bytes
                )
            self._events[slot].record(self._stream)

    def next(self) -> Dict[str, torch.Tensor]:
        """Return the prefetched batch and start gathering the next one."""
        if not self._primed:
            self._launch(self._cur)
            self._primed = True
        torch.cuda.current_stream().wait_event(self._events[self._cur])
        out = self._outs[self._cur]
        nxt = (self._cur + 1) % self._depth
        self._launch(nxt)
        self._cur = nxt
        return out

    def sync(self) -> None:
        """Block the host until in-flight gathers finish — call before any
        ``rb.add`` that may overwrite rows a pending gather could read."""
        self._stream.synchronize()
