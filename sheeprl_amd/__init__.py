"""sheeprl-amd: an MI355X-native distributed deep-RL framework.

A from-scratch framework with the capability set of Eclectic-Sheep/sheeprl
(13 algorithms: A2C, PPO[+decoupled/recurrent], SAC[+decoupled/AE], DroQ,
Dreamer-V1/V2/V3, Plan2Explore) re-designed for AMD Instinct MI355X:
PyTorch-ROCm as the tensor substrate, hand-written CDNA4 HIP kernels for the
hot ops, and an RCCL-over-xGMI runtime instead of Lightning Fabric.
"""

__version__ = "0.1.0"

import torch.distributions as _td

# Framework-wide default (parity: cfg.distribution.validate_args=False in the
# reference).  Validation constructs do a host-side `_is_all_true` per
# distribution — 232 stream syncs per DV3 train step, and they forbid hipGraph
# capture.  Re-enable per-run with distribution.validate_args=True.
_td.Distribution.set_default_validate_args(False)

from sheeprl_amd.utils.dotdict import DotDict  # noqa: F401
