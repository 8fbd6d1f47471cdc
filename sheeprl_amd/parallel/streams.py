"""Intra-graph concurrency: fork independent kernel chains onto side HIP
streams so a hipGraph capture records a DAG instead of a serial chain.

Why this exists on MI355X: the graphed Dreamer-V3 gradient step is ~1.8k
kernels of 5-9 us each on grids of 8-64 workgroups — a 256-CU chip executing
one tiny kernel at a time.  Captured on a single stream, replay wall time is
the *sum* of kernel times.  Sections of the step are mutually independent
(decoder vs reward vs continue heads over the same latents; the actor and
critic behaviour phases after imagination; the no-grad value/reward/continue
sweeps over imagined trajectories).  Running each such section inside a
``fork()`` context records it on its own stream, fenced from the ambient
stream with events, so hipGraph replay executes the branches concurrently on
idle CUs.

Autograd cooperation: the engine runs every op's backward on the stream that
ran its forward (and inserts the cross-stream events where gradients join),
so forking a section's *forward* parallelises its backward slice for free.

Allocator safety: PyTorch's caching allocator segregates blocks by
allocation stream, so a block freed inside one branch is never silently
reused by a concurrently-captured branch on another stream; branch outputs
consumed after ``join()`` are ordered by the join events.  Keep RNG and
collectives on the ambient stream (philox state capture and RCCL enqueue
order are single-stream concerns); the Dreamer sections forked here contain
neither.

Usage::

    br = Branches(enabled=use_streams)
    with br.fork():
        a = decoder(latents)      # records on side stream 0
    with br.fork():
        b = reward_head(latents)  # records on side stream 1 (concurrent)
    br.join()                     # ambient stream waits on both
    loss = combine(a, b)          # ordered after both branches

``enabled=False`` (or a CPU device) makes every call a no-op so the same
code path runs unchanged in eager CPU tests — the fork is a scheduling hint,
never a semantic change.
"""

from __future__ import annotations

from typing import List, Optional

import torch

__all__ = ["Branches"]

# Fixed global stream pool: warmup, capture and interleaved eager metric
# steps must all see the *same* stream objects, otherwise AccumulateGrad
# nodes record mismatched streams between warmup and capture.
_POOL: List["torch.cuda.Stream"] = []


def _side_stream(i: int) -> "torch.cuda.Stream":
    while len(_POOL) <= i:
        _POOL.append(torch.cuda.Stream())
    return _POOL[i]


class _Fork:
    """Context manager: run the body on side stream ``idx``, fenced after the
    ambient stream's current position."""

    def __init__(self, branches: "Branches", idx: int) -> None:
        self._branches = branches
        self._idx = idx
        self._ctx: Optional[object] = None

    def __enter__(self) -> "_Fork":
        if self._branches.enabled:
            s = _side_stream(self._idx)
            s.wait_stream(torch.cuda.current_stream())
            self._ctx = torch.cuda.stream(s)
            self._ctx.__enter__()
        return self

    def __exit__(self, *exc) -> None:
        if self._ctx is not None:
            self._ctx.__exit__(*exc)
            self._ctx = None


class Branches:
    """A fork/join region.  Each ``fork()`` call opens the next side stream;
    ``join()`` makes the ambient stream wait on every opened branch and
    resets the region for reuse."""

    def __init__(self, enabled: bool = True) -> None:
        self.enabled = bool(enabled) and torch.cuda.is_available()
        self._used = 0

    def fork(self) -> _Fork:
        f = _Fork(self, self._used)
        self._used += 1
        return f

    def join(self) -> None:
        if self.enabled:
            cur = torch.cuda.current_stream()
            for i in range(self._used):
                cur.wait_stream(_side_stream(i))
        self._used = 0
