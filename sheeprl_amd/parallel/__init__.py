from sheeprl_amd.parallel.runtime import Runtime, flat_to_params, get_single_device_runtime, params_to_flat
from sheeprl_amd.parallel.gradsync import GradSync

__all__ = ["Runtime", "GradSync", "get_single_device_runtime", "params_to_flat", "flat_to_params"]
