"""External environment backends (parity surface: sheeprl/envs/{dmc,minedojo,
minerl,diambra,crafter,super_mario_bros}.py).

None of the backing simulators ship in this offline image, so each backend
adapts the third-party env to the sheeprl-amd Env API **when its package is
importable** and raises an actionable error otherwise.  The adapters convert
HWC uint8 frames to the framework's channel-first convention and dict-ify
observations; everything else (action repeat, resize, frame stack, episode
stats) is composed by the standard wrapper pipeline in envs/factory.py.
"""

from __future__ import annotations

from typing import Any, Optional

import numpy as np

from sheeprl_amd.envs import spaces
from sheeprl_amd.envs.core import Env
from sheeprl_amd.envs.factory import register_env


class _GymnasiumAdapter(Env):
    """Adapts any gymnasium-API env instance (channel-last pixels)."""

    def __init__(self, env: Any, pixel_keys: tuple = ("rgb",)) -> None:
        self._env = env
        self._pixel_keys = pixel_keys
        self.observation_space = self._convert_obs_space(env.observation_space)
        self.action_space = self._convert_act_space(env.action_space)

    @staticmethod
    def _to_chw(x: np.ndarray) -> np.ndarray:
        return np.transpose(x, (2, 0, 1)) if x.ndim == 3 else x

    def _convert_obs_space(self, sp: Any) -> spaces.Space:
        if hasattr(sp, "spaces"):  # dict space
            return spaces.Dict({k: self._convert_obs_space(v) for k, v in sp.spaces.items()})
        shape = tuple(sp.shape)
        if len(shape) == 3:  # HWC pixels -> CHW
            return spaces.Box(0, 255, (shape[2], shape[0], shape[1]), np.uint8)
        return spaces.Box(np.asarray(sp.low), np.asarray(sp.high), shape, sp.dtype)

    def _convert_act_space(self, sp: Any) -> spaces.Space:
        if hasattr(sp, "n"):
            return spaces.Discrete(int(sp.n))
        if hasattr(sp, "nvec"):
            return spaces.MultiDiscrete([int(n) for n in sp.nvec])
        return spaces.Box(np.asarray(sp.low), np.asarray(sp.high), tuple(sp.shape), np.float32)

    def _convert_obs(self, obs: Any) -> Any:
        if isinstance(obs, dict):
            return {k: self._to_chw(np.asarray(v)) for k, v in obs.items()}
        arr = np.asarray(obs)
        return self._to_chw(arr)

    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None):
        obs, info = self._env.reset(seed=seed, options=options)
        return self._convert_obs(obs), dict(info)

    def step(self, action: Any):
        obs, r, term, trunc, info = self._env.step(action)
        return self._convert_obs(obs), float(r), bool(term), bool(trunc), dict(info)

    def close(self) -> None:
        self._env.close()


def _missing(name: str, package: str):
    def builder(**kwargs: Any) -> Env:
        raise ImportError(
            f"env backend '{name}' needs the '{package}' package, which is not installed in this image. "
            f"Install it and the backend activates automatically."
        )

    return builder


def _make_dmc(domain: str = "walker", task: str = "walk", **kwargs: Any) -> Env:
    """DeepMind Control via dm_control (parity: sheeprl/envs/dmc.py)."""
    try:
        from dm_control import suite  # type: ignore
    except ImportError as e:
        raise ImportError("env backend 'dmc' needs dm_control (not in this image)") from e

    dmc_env = suite.load(domain, task)

    class DMCEnv(Env):
        def __init__(self) -> None:
            spec = dmc_env.action_spec()
            self.action_space = spaces.Box(spec.minimum, spec.maximum, tuple(spec.shape), np.float32)
            obs_spec = dmc_env.observation_spec()
            self.observation_space = spaces.Dict(
                {k: spaces.Box(-np.inf, np.inf, tuple(v.shape) or (1,), np.float32) for k, v in obs_spec.items()}
            )

        def reset(self, *, seed=None, options=None):
            ts = dmc_env.reset()
            return {k: np.asarray(v, np.float32).reshape(-1) for k, v in ts.observation.items()}, {}

        def step(self, action):
            ts = dmc_env.step(action)
            obs = {k: np.asarray(v, np.float32).reshape(-1) for k, v in ts.observation.items()}
            return obs, float(ts.reward or 0.0), ts.last(), False, {}

    return DMCEnv()


def _make_crafter(**kwargs: Any) -> Env:
    """Crafter (parity: sheeprl/envs/crafter.py)."""
    try:
        import crafter  # type: ignore
    except ImportError as e:
        raise ImportError("env backend 'crafter' needs the crafter package (not in this image)") from e
    return _GymnasiumAdapter(crafter.Env(**kwargs))


def _make_atari(env_id: str = "MsPacmanNoFrameskip-v4", **kwargs: Any) -> Env:
    """ALE Atari via gymnasium (parity: the reference's env=atari config)."""
    try:
        import gymnasium  # type: ignore
    except ImportError as e:
        raise ImportError("env backend 'atari' needs gymnasium[atari] (not in this image); "
                          "use id=synthetic_atari for offline benchmarking") from e
    return _GymnasiumAdapter(gymnasium.make(env_id, render_mode="rgb_array"))


def _make_gym(env_id: str, **kwargs: Any) -> Env:
    """Generic gymnasium env (the reference's env=gym / env=mujoco configs)."""
    try:
        import gymnasium  # type: ignore
    except ImportError as e:
        raise ImportError("env backend 'gym' needs gymnasium (not in this image)") from e
    return _GymnasiumAdapter(gymnasium.make(env_id, render_mode="rgb_array"))


register_env("gym", _make_gym)
register_env("mujoco", _make_gym)
register_env("dmc", _make_dmc)
register_env("crafter", _make_crafter)
register_env("atari", _make_atari)
def _make_minedojo(**kwargs: Any) -> Env:
    from sheeprl_amd.envs.minedojo_adapter import MineDojoAdapter

    return MineDojoAdapter(**kwargs)


def _make_minerl(**kwargs: Any) -> Env:
    from sheeprl_amd.envs.minerl_adapter import MineRLAdapter

    return MineRLAdapter(**kwargs)


def _make_diambra(**kwargs: Any) -> Env:
    from sheeprl_amd.envs.diambra_adapter import DiambraAdapter

    return DiambraAdapter(**kwargs)


# full adapters (action flattening / masks / settings plumbing implemented and
# unit-tested against fakes); they raise an actionable ImportError when the
# backing simulator package is absent from the image
register_env("minedojo", _make_minedojo)
register_env("minerl", _make_minerl)
register_env("diambra", _make_diambra)
register_env("super_mario_bros", _missing("super_mario_bros", "gym-super-mario-bros"))
