"""DIAMBRA arena adapter (parity surface: sheeprl/envs/diambra.py:20-145).

Settings plumbing + observation normalization for the DIAMBRA fighting-game
arena: discrete/multi-discrete scalar observations become ``Box`` vectors so
the encoder pipeline can consume them, the frame shape rides the engine-side
settings (``increase_performance``) or the wrapper side, sticky actions force
``step_ratio=1``, and the engine's ``env_done`` flag folds into termination.

The ``diambra`` package is not in this image, so the arena module is
injectable (``backend=``) and the plumbing is unit-tested against a fake
(tests/test_envs_adapters.py); with the real package installed the adapter
binds to ``diambra.arena``.
"""

from __future__ import annotations

import warnings
from typing import Any, Dict, Optional, Tuple, Union

import numpy as np

from sheeprl_amd.envs import spaces
from sheeprl_amd.envs.core import Env


class DiambraAdapter(Env):
    def __init__(
        self,
        id: str,
        action_space: str = "DISCRETE",
        screen_size: Union[int, Tuple[int, int]] = 64,
        grayscale: bool = False,
        repeat_action: int = 1,
        rank: int = 0,
        diambra_settings: Optional[Dict[str, Any]] = None,
        diambra_wrappers: Optional[Dict[str, Any]] = None,
        render_mode: str = "rgb_array",
        log_level: int = 0,
        increase_performance: bool = True,
        backend: Any = None,  # injectable `diambra.arena`-like module (tests)
    ) -> None:
        if backend is None:
            try:
                import diambra.arena  # noqa: PLC0415

                backend = diambra.arena
            except ImportError as e:  # pragma: no cover
                raise ImportError(
                    "diambra is not installed; `pip install diambra diambra-arena` "
                    "and run under the diambra CLI, or pass a backend module"
                ) from e
        if isinstance(screen_size, int):
            screen_size = (screen_size, screen_size)
        if action_space not in ("DISCRETE", "MULTI_DISCRETE"):
            raise ValueError(f"action_space must be DISCRETE or MULTI_DISCRETE, got {action_space}")
        settings = dict(diambra_settings or {})
        wrappers = dict(diambra_wrappers or {})
        for bad in ("frame_shape", "n_players"):
            if settings.pop(bad, None) is not None:
                warnings.warn(f"the DIAMBRA {bad} setting is managed by the adapter and was dropped")
        role = settings.pop("role", None)
        if role is not None and role not in ("P1", "P2"):
            raise ValueError(f"role must be P1, P2 or None, got {role}")
        self._discrete_actions = action_space == "DISCRETE"
        settings.update(
            game_id=id,
            action_space=getattr(backend.SpaceTypes, action_space),
            n_players=1,
            role=getattr(backend.Roles, role) if role is not None else None,
            render_mode=render_mode,
        )
        if repeat_action > 1:
            if settings.get("step_ratio", 6) > 1:
                warnings.warn(f"step_ratio forced to 1: sticky actions active ({repeat_action})")
            settings["step_ratio"] = 1
        for bad in ("frame_shape", "stack_frames", "dilation", "flatten"):
            if wrappers.pop(bad, None) is not None:
                warnings.warn(f"the DIAMBRA {bad} wrapper is managed by the adapter and was dropped")
        wrappers.update(flatten=True, repeat_action=repeat_action)
        frame_shape = (*screen_size, int(grayscale))
        if increase_performance:
            settings["frame_shape"] = frame_shape  # engine-side resize (fast)
        else:
            wrappers["frame_shape"] = frame_shape  # wrapper-side resize
        self._env = backend.make(id, settings, wrappers, rank=rank,
                                 render_mode=render_mode, log_level=log_level)

        self.action_space = self._convert_act_space(self._env.action_space)
        obs_spaces: Dict[str, spaces.Space] = {}
        for k, sp in self._env.observation_space.spaces.items():
            if hasattr(sp, "n"):  # Discrete scalar -> 1-dim Box
                obs_spaces[k] = spaces.Box(0, int(sp.n) - 1, (1,), np.int32)
            elif hasattr(sp, "nvec"):  # MultiDiscrete -> vector Box
                nv = np.asarray(sp.nvec)
                obs_spaces[k] = spaces.Box(np.zeros_like(nv), nv - 1, (len(nv),), np.int32)
            elif hasattr(sp, "shape"):
                obs_spaces[k] = spaces.Box(np.asarray(sp.low), np.asarray(sp.high), tuple(sp.shape), sp.dtype)
            else:
                raise RuntimeError(f"unsupported DIAMBRA observation space for {k}: {type(sp)}")
        self.observation_space = spaces.Dict(obs_spaces)
        self.render_mode = render_mode

    @staticmethod
    def _convert_act_space(sp: Any) -> spaces.Space:
        if hasattr(sp, "n"):
            return spaces.Discrete(int(sp.n))
        return spaces.MultiDiscrete([int(n) for n in sp.nvec])

    def _convert_obs(self, obs: Dict[str, Any]) -> Dict[str, np.ndarray]:
        return {
            k: np.asarray(v).reshape(self.observation_space[k].shape)
            for k, v in obs.items()
        }

    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None):
        obs, info = self._env.reset(seed=seed, options=options)
        info = dict(info)
        info["env_domain"] = "DIAMBRA"
        return self._convert_obs(obs), info

    def step(self, action: Any):
        if self._discrete_actions and isinstance(action, np.ndarray):
            action = int(np.asarray(action).squeeze().item())
        obs, reward, terminated, truncated, info = self._env.step(action)
        info = dict(info)
        info["env_domain"] = "DIAMBRA"
        return (
            self._convert_obs(obs),
            float(reward),
            bool(terminated) or bool(info.get("env_done", False)),
            bool(truncated),
            info,
        )

    def render(self):
        return self._env.render()

    def close(self) -> None:
        self._env.close()
