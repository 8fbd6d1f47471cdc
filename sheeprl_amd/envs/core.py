"""Base environment API (gymnasium-compatible step/reset signatures).

``reset(seed=..., options=...) -> (obs, info)``;
``step(action) -> (obs, reward, terminated, truncated, info)``.
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

from sheeprl_amd.envs import spaces


class Env:
    observation_space: spaces.Space
    action_space: spaces.Space
    metadata: Dict[str, Any] = {"render_fps": 30}
    render_mode: Optional[str] = None
    reward_range: Tuple[float, float] = (-float("inf"), float("inf"))

    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None) -> Tuple[Any, dict]:
        raise NotImplementedError

    def step(self, action: Any) -> Tuple[Any, float, bool, bool, dict]:
        raise NotImplementedError

    def render(self) -> Any:
        return None

    def close(self) -> None:
        pass

    @property
    def unwrapped(self) -> "Env":
        return self

    def __enter__(self) -> "Env":
        return self

    def __exit__(self, *exc: Any) -> None:
        self.close()


class Wrapper(Env):
    def __init__(self, env: Env) -> None:
        self.env = env
        self.observation_space = env.observation_space
        self.action_space = env.action_space
        self.metadata = env.metadata
        self.render_mode = env.render_mode

    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None) -> Tuple[Any, dict]:
        return self.env.reset(seed=seed, options=options)

    def step(self, action: Any) -> Tuple[Any, float, bool, bool, dict]:
        return self.env.step(action)

    def render(self) -> Any:
        return self.env.render()

    def close(self) -> None:
        self.env.close()

    @property
    def unwrapped(self) -> Env:
        return self.env.unwrapped

    def __getattr__(self, name: str) -> Any:
        if name.startswith("_"):
            raise AttributeError(name)
        return getattr(self.env, name)


class ObservationWrapper(Wrapper):
    def observation(self, obs: Any) -> Any:
        raise NotImplementedError

    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None) -> Tuple[Any, dict]:
        obs, info = self.env.reset(seed=seed, options=options)
        return self.observation(obs), info

    def step(self, action: Any) -> Tuple[Any, float, bool, bool, dict]:
        obs, r, term, trunc, info = self.env.step(action)
        return self.observation(obs), r, term, trunc, info


class ActionWrapper(Wrapper):
    def action(self, action: Any) -> Any:
        raise NotImplementedError

    def step(self, action: Any) -> Tuple[Any, float, bool, bool, dict]:
        return self.env.step(self.action(action))


class RewardWrapper(Wrapper):
    def reward(self, r: float) -> float:
        raise NotImplementedError

    def step(self, action: Any) -> Tuple[Any, float, bool, bool, dict]:
        obs, r, term, trunc, info = self.env.step(action)
        return obs, self.reward(r), term, trunc, info
