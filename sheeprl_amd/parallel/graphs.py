"""hipGraph capture of whole training iterations.

The Dreamer training step is ~20k tiny kernels per iteration (sequential RSSM
scan over T=64 plus imagination over H=15); eager dispatch makes it
host-bound on MI355X (measured: 204 ms CPU vs 138 ms GPU per step).  Capturing
forward + backward + optimizer step in a single hipGraph collapses the host
cost to one graph launch.

Capture-safety requirements handled elsewhere in the framework:
* RNG through philox ops only (gumbel-max categorical sampling,
  sheeprl_amd/distributions/dists.py:gumbel_onehot_sample);
* optimizer bias correction from a device step scalar
  (FusedAdam + adam_step_dev kernel);
* cross-step state (Moments EMA, Adam moments, EMA target update) mutated
  in place on fixed storage.
"""

from __future__ import annotations

from typing import Callable, Dict, Optional

import torch


class CUDAGraphStep:
    """Capture ``fn(static_inputs)`` once, then replay with new input values.

    ``example_inputs`` is a dict of tensors; their clones become the static
    buffers.  Warmup iterations run on a side stream first (allocator +
    MIOpen/hipBLASLt find must be settled before capture).
    """

    def __init__(
        self,
        fn: Callable[[Dict[str, torch.Tensor]], None],
        example_inputs: Dict[str, torch.Tensor],
        warmup: int = 3,
        pool: Optional[object] = None,
    ) -> None:
        self.fn = fn
        self.static: Dict[str, torch.Tensor] = {
            k: v.detach().clone() if isinstance(v, torch.Tensor) else v for k, v in example_inputs.items()
        }
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup):
                fn(self.static)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        # thread_local error mode: RCCL's proxy threads issue HIP calls of
        # their own during capture; global mode would abort the capture
        with torch.cuda.graph(self.graph, pool=pool, capture_error_mode="thread_local"):
            fn(self.static)

    def __call__(self, inputs: Dict[str, torch.Tensor]) -> None:
        for k, v in inputs.items():
            self.static[k].copy_(v, non_blocking=True)
        self.graph.replay()
