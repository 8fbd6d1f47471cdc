from sheeprl_amd.parallel.runtime import Runtime, get_single_device_runtime
from sheeprl_amd.parallel.gradsync import GradSync

__all__ = ["Runtime", "GradSync", "get_single_device_runtime"]
