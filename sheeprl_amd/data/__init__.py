from sheeprl_amd.data.buffers import (
    EnvIndependentReplayBuffer,
    EpisodeBuffer,
    ReplayBuffer,
    SequentialReplayBuffer,
    get_tensor,
)
from sheeprl_amd.data.memmap import MemmapArray

__all__ = [
    "ReplayBuffer",
    "SequentialReplayBuffer",
    "EnvIndependentReplayBuffer",
    "EpisodeBuffer",
    "MemmapArray",
    "get_tensor",
]
