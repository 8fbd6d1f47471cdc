"""Env factory: config -> thunk building a fully-wrapped environment.

Parity with sheeprl/utils/env.py:26-231 (``make_env``): action repeat,
velocity masking, dict-ification of flat observations, grayscale/resize,
frame stacking, actions/reward-as-observation, time limit, episode stats.
Env backends are the in-repo ones (no gym / ALE in this image): dummy_*,
cartpole, pendulum, synthetic_atari.
"""

from __future__ import annotations

from typing import Any, Callable, Dict, Optional

from sheeprl_amd.envs import spaces, wrappers
from sheeprl_amd.envs.classic import CartPoleEnv, PendulumEnv
from sheeprl_amd.envs.core import Env
from sheeprl_amd.envs.dummy import ContinuousDummyEnv, DiscreteDummyEnv, MultiDiscreteDummyEnv
from sheeprl_amd.envs.synthetic import SyntheticAtariEnv
from sheeprl_amd.envs.vector import AsyncVectorEnv, SyncVectorEnv

_ENV_BUILDERS: Dict[str, Callable[..., Env]] = {
    "dummy_continuous": ContinuousDummyEnv,
    "dummy_discrete": DiscreteDummyEnv,
    "dummy_multidiscrete": MultiDiscreteDummyEnv,
    "cartpole": CartPoleEnv,
    "CartPole-v1": CartPoleEnv,
    "pendulum": PendulumEnv,
    "Pendulum-v1": PendulumEnv,
    "synthetic_atari": SyntheticAtariEnv,
}


def register_env(env_id: str, builder: Callable[..., Env]) -> None:
    _ENV_BUILDERS[env_id] = builder


def make_env(
    cfg: Any,
    seed: int,
    rank: int = 0,
    run_name: Optional[str] = None,
    prefix: str = "",
    vector_env_idx: int = 0,
) -> Callable[[], Env]:
    """Returns a thunk creating one fully-wrapped env instance."""

    def thunk() -> Env:
        env_cfg = cfg.env
        env_id = env_cfg.id
        kwargs = dict(env_cfg.get("wrapper_kwargs", {}) or {})
        if env_id in _ENV_BUILDERS:
            env: Env = _ENV_BUILDERS[env_id](**kwargs)
        else:
            # route by explicit backend (env/atari.yaml etc.) or id prefix to
            # the external-package adapters (sheeprl_amd/envs/external.py)
            backend = env_cfg.get("backend", None)
            if backend is None:
                pfx = {
                    "dmc_": "dmc", "crafter_": "crafter", "minedojo_": "minedojo",
                    "minerl_": "minerl", "diambra_": "diambra", "supermario_": "super_mario_bros",
                }
                for k, v in pfx.items():
                    if str(env_id).startswith(k):
                        backend = v
                        break
                if backend is None and ("NoFrameskip" in str(env_id) or str(env_id).startswith("ALE/")):
                    backend = "atari"
            if backend is None or backend not in _ENV_BUILDERS:
                raise ValueError(f"unknown env id '{env_id}'; known: {sorted(_ENV_BUILDERS)}")
            env = _ENV_BUILDERS[backend](env_id, **kwargs)

        if env_cfg.get("mask_velocities", False):
            env = wrappers.MaskVelocityWrapper(env, env_id)
        if env_cfg.get("action_repeat", 1) > 1:
            env = wrappers.ActionRepeat(env, env_cfg.action_repeat)
        env = wrappers.DictObservation(env, key="state")

        cnn_keys = list(cfg.algo.cnn_keys.encoder or []) if "algo" in cfg else []
        if env_cfg.get("grayscale", False) and cnn_keys:
            env = wrappers.GrayscaleObservation(env, cnn_keys)
        screen = env_cfg.get("screen_size", None)
        if screen and cnn_keys and any(
            k in env.observation_space.spaces and env.observation_space[k].shape[1:] != (screen, screen)
            for k in cnn_keys
        ):
            env = wrappers.ResizeObservation(env, (screen, screen), cnn_keys)
        if env_cfg.get("frame_stack", 1) > 1 and cnn_keys:
            env = wrappers.FrameStack(env, env_cfg.frame_stack, cnn_keys, env_cfg.get("frame_stack_dilation", 1))
        if env_cfg.get("reward_as_observation", False):
            env = wrappers.RewardAsObservation(env)
        aao = env_cfg.get("actions_as_observation", None)
        if aao and aao.get("num_stack", -1) > 0:
            env = wrappers.ActionsAsObservation(env, aao["num_stack"], aao["noop"], aao.get("dilation", 1))
        if env_cfg.get("clip_rewards", False):
            env = wrappers.ClipReward(env)
        if env_cfg.get("max_episode_steps") and env_cfg.max_episode_steps > 0:
            env = wrappers.TimeLimit(env, env_cfg.max_episode_steps)
        if env_cfg.get("capture_video", False) and run_name and rank == 0 and vector_env_idx == 0 and cnn_keys:
            env = wrappers.RecordVideo(env, f"logs/runs/{run_name}/videos", key=cnn_keys[0])
        env = wrappers.RecordEpisodeStatistics(env)

        env.observation_space.seed(seed + rank * 1024 + vector_env_idx)
        env.action_space.seed(seed + rank * 1024 + vector_env_idx)
        return env

    return thunk


def vectorize_env(cfg: Any, seed: int, rank: int, run_name: Optional[str] = None) -> Any:
    fns = [
        make_env(cfg, seed + rank * cfg.env.num_envs + i, rank, run_name, vector_env_idx=i)
        for i in range(cfg.env.num_envs)
    ]
    if cfg.env.get("sync_env", True):
        return SyncVectorEnv(fns)
    return AsyncVectorEnv(fns)
