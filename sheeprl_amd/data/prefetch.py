"""Pinned-host replay prefetching (BASELINE.json north star: "the numpy
replay buffer becomes a pinned-host ring with hipMemcpyAsync prefetch into
HBM so minibatch sampling overlaps world-model training on a side HIP
stream").

A background thread gathers the next minibatch from the (numpy) replay
buffer into PINNED staging tensors while the GPU is still executing the
current gradient step; the H2D copies are issued with ``non_blocking=True``
on a dedicated side stream, and the consumer's stream waits on a recorded
event — the copy itself rides hipMemcpyAsync over a different queue than the
compute, so sampling disappears from the critical path.
"""

from __future__ import annotations

import queue
import threading
from typing import Callable, Dict, Optional

import torch


class DevicePrefetcher:
    """Wraps a ``sample() -> dict[str, np.ndarray-like torch tensors]``
    callable; ``next()`` returns device batches one step ahead."""

    def __init__(
        self,
        sample_fn: Callable[[], Dict[str, torch.Tensor]],
        device: torch.device,
        depth: int = 2,
        lock: Optional[threading.Lock] = None,
    ) -> None:
        self.sample_fn = sample_fn
        self.lock = lock or threading.Lock()
        self.device = device
        self.use_cuda = device.type == "cuda"
        self.stream = torch.cuda.Stream() if self.use_cuda else None
        self._q: "queue.Queue" = queue.Queue(maxsize=depth)
        self._stop = threading.Event()
        self._err: Optional[BaseException] = None
        # ring of pinned staging sets: a slot is only reused after depth+2
        # batches, by which time its async H2D must have been consumed — a
        # single set would let the worker overwrite host memory still being
        # copied by a previous non_blocking transfer
        self._pinned_ring = [dict() for _ in range(depth + 2)]
        self._ring_idx = 0
        self._thread = threading.Thread(target=self._worker, daemon=True)
        self._thread.start()

    def _stage(self, host_batch: Dict[str, torch.Tensor]):
        if not self.use_cuda:
            return host_batch, None
        out: Dict[str, torch.Tensor] = {}
        pinned = self._pinned_ring[self._ring_idx]
        self._ring_idx = (self._ring_idx + 1) % len(self._pinned_ring)
        with torch.cuda.stream(self.stream):
            for k, v in host_batch.items():
                pin = pinned.get(k)
                if pin is None or pin.shape != v.shape or pin.dtype != v.dtype:
                    pin = torch.empty_like(v, pin_memory=True)
                    pinned[k] = pin
                pin.copy_(v)
                out[k] = pin.to(self.device, non_blocking=True)
            event = torch.cuda.Event()
            event.record(self.stream)
        return out, event

    def _worker(self) -> None:
        try:
            while not self._stop.is_set():
                with self.lock:  # buffer writers share this lock (no torn rows)
                    host = self.sample_fn()
                staged = self._stage(host)
                while not self._stop.is_set():
                    try:
                        self._q.put(staged, timeout=0.1)
                        break
                    except queue.Full:
                        continue
        except BaseException as e:  # noqa: BLE001
            self._err = e
            self._q.put(None)

    def next(self) -> Dict[str, torch.Tensor]:
        item = self._q.get()
        if item is None:
            raise RuntimeError(f"prefetch worker died: {self._err}") from self._err
        batch, event = item
        if event is not None:
            torch.cuda.current_stream().wait_event(event)
        return batch

    def close(self) -> None:
        self._stop.set()
        try:
            while True:
                self._q.get_nowait()
        except queue.Empty:
            pass
        self._thread.join(timeout=5)
