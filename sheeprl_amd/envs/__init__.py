from sheeprl_amd.envs import spaces
from sheeprl_amd.envs.core import Env, Wrapper, ObservationWrapper, ActionWrapper, RewardWrapper
from sheeprl_amd.envs.factory import make_env, vectorize_env, register_env
from sheeprl_amd.envs.vector import SyncVectorEnv, AsyncVectorEnv
from sheeprl_amd.envs import external  # noqa: F401  (registers external backends)

__all__ = [
    "spaces",
    "Env",
    "Wrapper",
    "ObservationWrapper",
    "ActionWrapper",
    "RewardWrapper",
    "make_env",
    "vectorize_env",
    "register_env",
    "SyncVectorEnv",
    "AsyncVectorEnv",
]
