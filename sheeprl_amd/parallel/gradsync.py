"""Bucketed gradient all-reduce with backward overlap (the DDP replacement).

Replaces the implicit DDP gradient hooks of the reference's
``fabric.setup_module`` ("implicit DDP gradient all-reduce on
every fabric.backward").  Design targets RCCL over xGMI:

* buckets are large (64 MiB default) because each ring step is bound by one
  xGMI link (~153 GB/s); many small buckets pay latency, not bandwidth.
* each bucket owns a persistent flat buffer; when the last gradient of a
  bucket materialises during backward, the bucket is flattened and an async
  ``all_reduce`` is launched immediately — RCCL runs it on its own stream so
  communication overlaps the rest of backward.
* ``finalize()`` (called by ``Runtime.backward``) launches any straggler
  buckets (zero-filling params that produced no grad this pass — this is the
  find-unused-parameters case the reference needs for SAC-AE, cli.py:108-116),
  waits for all handles and writes the averaged gradients back.
* buckets are built in reverse parameter order, approximating autograd's
  execution order so early buckets fill early.
"""

from __future__ import annotations

from typing import Any, List, Optional

import torch
import torch.distributed as dist
from torch import nn


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter], dtype: torch.dtype, device: torch.device) -> None:
        self.params = params
        self.numels = [p.numel() for p in params]
        self.offsets: List[int] = []
        off = 0
        for n in self.numels:
            self.offsets.append(off)
            off += n
        self.flat = torch.zeros(off, dtype=dtype, device=device)
        self.ready = 0
        self.seen = [False] * len(params)
        self.handle: Optional[Any] = None
        self.index = {id(p): i for i, p in enumerate(params)}

    def mark_ready(self, p: torch.nn.Parameter) -> bool:
        i = self.index[id(p)]
        if not self.seen[i]:
            self.seen[i] = True
            self.ready += 1
        return self.ready == len(self.params)

    def launch(self, group: Any, world_size: int) -> None:
        for i, p in enumerate(self.params):
            dst = self.flat[self.offsets[i] : self.offsets[i] + self.numels[i]]
            if self.seen[i] and p.grad is not None:
                # reshape, not view: channels_last conv grads have permuted
                # strides and are not viewable as 1-D
                dst.copy_(p.grad.detach().reshape(-1))
            else:
                dst.zero_()
        self.flat.div_(world_size)
        self.handle = dist.all_reduce(self.flat, group=group, async_op=True)

    def finish(self) -> None:
        if self.handle is not None:
            self.handle.wait()
            self.handle = None
        for i, p in enumerate(self.params):
            src = self.flat[self.offsets[i] : self.offsets[i] + self.numels[i]].view(p.shape)
            if p.grad is None:
                # empty_like preserves the param's memory format so the grad
                # layout matches (keeps fused-optimizer fast paths eligible)
                p.grad = torch.empty_like(p).copy_(src)
            else:
                p.grad.detach().copy_(src)
        self.ready = 0
        self.seen = [False] * len(self.params)


class GradSync:
    def __init__(
        self,
        module: nn.Module,
        bucket_cap_mb: int = 64,
        process_group: Any = None,
        world_size: Optional[int] = None,
    ) -> None:
        self.module = module
        self.group = process_group
        self.world_size = world_size or dist.get_world_size(process_group)
        self.enabled = True
        self._any_ready = False
        self._launch_order: List[int] = []  # bucket launch order of the last real round
        self._any_launched = False

        params = [p for p in module.parameters() if p.requires_grad]
        cap = bucket_cap_mb * 1024 * 1024
        self.buckets: List[_Bucket] = []
        self._param_bucket = {}
        cur: List[torch.nn.Parameter] = []
        cur_bytes = 0
        cur_dtype: Optional[torch.dtype] = None
        # reverse order ~ backward execution order
        for p in reversed(params):
            b = p.numel() * p.element_size()
            if cur and (cur_bytes + b > cap or p.dtype != cur_dtype):
                self._commit(cur, cur_dtype)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += b
            cur_dtype = p.dtype
        if cur:
            self._commit(cur, cur_dtype)

        for p in params:
            p.register_post_accumulate_grad_hook(self._hook)

    def _commit(self, params: List[torch.nn.Parameter], dtype: torch.dtype) -> None:
        bucket = _Bucket(params, dtype, params[0].device)
        self.buckets.append(bucket)
        for p in params:
            self._param_bucket[id(p)] = bucket

    def _hook(self, p: torch.nn.Parameter) -> None:
        if not self.enabled or not dist.is_initialized():
            return
        self._any_ready = True
        bucket = self._param_bucket[id(p)]
        if bucket.mark_ready(p):
            if not self._any_launched:
                self._launch_order = []
                self._any_launched = True
            self._launch_order.append(self.buckets.index(bucket))
            bucket.launch(self.group, self.world_size)

    def finalize(self) -> None:
        """Launch stragglers, wait for all in-flight reduces, write back.

        Collective matching relies on data-INDEPENDENT control flow: every
        rank runs the same module graph per step (true for all shipped
        algorithms), so the same buckets fire in the same order everywhere.
        """
        if not self.enabled or not self._any_ready:
            return
        for i, b in enumerate(self.buckets):
            if b.handle is None and b.ready > 0:
                self._launch_order.append(i)
                b.launch(self.group, self.world_size)
        for b in self.buckets:
            if b.handle is not None:
                b.finish()
        self._any_ready = False
        self._any_launched = False

    def sync_zero(self) -> None:
        """Join-equivalent shadow round for a trainer that ran out of local
        minibatches (uneven rollout chunks, reference ppo_decoupled.py:497-499):
        launches every bucket's all-reduce with ZERO local gradients, in the
        same bucket order the peers' backward produces, waits, and writes the
        averaged peer gradients back.  Followed by optimizer.step() this keeps
        every replica bit-identical (unlike torch's Join, which leaves joined
        ranks stale)."""
        if not self.enabled or not dist.is_initialized():
            return
        # replay the last real round's bucket launch order so the zero round's
        # collectives match the active peers' issue order
        order = self._launch_order if len(self._launch_order) == len(self.buckets) else list(range(len(self.buckets)))
        for i in order:
            b = self.buckets[i]
            b.flat.zero_()
            b.handle = dist.all_reduce(b.flat, group=self.group, async_op=True)
        for b in self.buckets:
            b.finish()

    def broadcast_params(self, src: int = 0) -> None:
        """One flat broadcast per dtype instead of per-tensor (init-time sync;
        also layout-safe: flattening copies channels_last tensors densely)."""
        with torch.no_grad():
            by_dtype = {}
            for p in self.module.parameters():
                by_dtype.setdefault(p.dtype, []).append(p.data)
            for b in self.module.buffers():
                if b.dtype.is_floating_point or b.dtype in (torch.int32, torch.int64, torch.uint8, torch.bool):
                    by_dtype.setdefault(b.dtype, []).append(b.data)
            for ts in by_dtype.values():
                flat = torch.cat([t.reshape(-1) for t in ts])
                dist.broadcast(flat, src=src, group=self.group)
                off = 0
                for t in ts:
                    n = t.numel()
                    t.copy_(flat[off : off + n].view(t.shape))
                    off += n
