"""Environment wrappers.

Parity with sheeprl/envs/wrappers.py: MaskVelocityWrapper
(:13), ActionRepeat (:48), RestartOnException (:74), FrameStack (:126),
RewardAsObservationWrapper (:185), ActionsAsObservationWrapper (:258) —
plus the stock gymnasium wrappers the reference composes in
sheeprl/utils/env.py (TimeLimit, RecordEpisodeStatistics, ClipReward,
grayscale/resize transforms).
"""

from __future__ import annotations

import time
from collections import deque
from typing import Any, Callable, Dict, Tuple

import numpy as np

from sheeprl_amd.envs import spaces
from sheeprl_amd.envs.core import Env, ObservationWrapper, Wrapper


class TimeLimit(Wrapper):
    def __init__(self, env: Env, max_episode_steps: int) -> None:
        super().__init__(env)
        self._max = int(max_episode_steps)
        self._elapsed = 0

    def reset(self, *, seed=None, options=None):
        self._elapsed = 0
        return self.env.reset(seed=seed, options=options)

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        self._elapsed += 1
        if self._elapsed >= self._max and not term:
            trunc = True
        return obs, r, term, trunc, info


class ActionRepeat(Wrapper):
    """Repeat each action ``amount`` times, summing rewards
    (parity: sheeprl/envs/wrappers.py:48)."""

    def __init__(self, env: Env, amount: int) -> None:
        super().__init__(env)
        if amount <= 0:
            raise ValueError("amount must be > 0")
        self.amount = int(amount)

    @property
    def action_repeat(self) -> int:
        return self.amount

    def step(self, action):
        total = 0.0
        term = trunc = False
        obs, info = None, {}
        for _ in range(self.amount):
            obs, r, term, trunc, info = self.env.step(action)
            total += float(r)
            if term or trunc:
                break
        return obs, total, term, trunc, info


class RecordEpisodeStatistics(Wrapper):
    """Adds ``info["episode"] = {"r": return, "l": length, "t": seconds}``
    on episode end (gymnasium convention the reference logs from)."""

    def __init__(self, env: Env) -> None:
        super().__init__(env)
        self._ret = 0.0
        self._len = 0
        self._t0 = time.perf_counter()

    def reset(self, *, seed=None, options=None):
        self._ret, self._len, self._t0 = 0.0, 0, time.perf_counter()
        return self.env.reset(seed=seed, options=options)

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        self._ret += float(r)
        self._len += 1
        if term or trunc:
            info = dict(info)
            info["episode"] = {
                "r": np.array([self._ret], dtype=np.float32),
                "l": np.array([self._len], dtype=np.int64),
                "t": np.array([time.perf_counter() - self._t0], dtype=np.float32),
            }
        return obs, r, term, trunc, info


class ClipReward(Wrapper):
    def __init__(self, env: Env, low: float = -1.0, high: float = 1.0) -> None:
        super().__init__(env)
        self.low, self.high = low, high

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        return obs, float(np.clip(r, self.low, self.high)), term, trunc, info


class MaskVelocityWrapper(ObservationWrapper):
    """Zero out velocity entries of classic-control observations
    (parity: sheeprl/envs/wrappers.py:13)."""

    _masks: Dict[str, np.ndarray] = {
        "CartPole-v1": np.array([1.0, 0.0, 1.0, 0.0], dtype=np.float32),
        "Pendulum-v1": np.array([1.0, 1.0, 0.0], dtype=np.float32),
    }

    def __init__(self, env: Env, env_id: str) -> None:
        super().__init__(env)
        if env_id not in self._masks:
            raise NotImplementedError(f"velocity masking not defined for {env_id}")
        self.mask = self._masks[env_id]

    def observation(self, obs):
        return obs * self.mask


class FrameStack(ObservationWrapper):
    """Stack the last ``num_stack`` image observations of every cnn key, with
    optional temporal dilation (parity: sheeprl/envs/wrappers.py:126)."""

    def __init__(self, env: Env, num_stack: int, cnn_keys: list, dilation: int = 1) -> None:
        super().__init__(env)
        if num_stack <= 0:
            raise ValueError(f"num_stack must be > 0, got {num_stack}")
        if not isinstance(env.observation_space, spaces.Dict):
            raise RuntimeError(f"FrameStack requires dict observations, got {env.observation_space}")
        self.num_stack = int(num_stack)
        self.dilation = int(dilation)
        self._cnn_keys = [
            k for k in (cnn_keys or []) if k in env.observation_space.spaces and len(env.observation_space[k].shape) == 3
        ]
        if not self._cnn_keys:
            raise RuntimeError(f"no valid cnn keys for frame stacking among {cnn_keys}")
        self._frames: Dict[str, deque] = {k: deque(maxlen=num_stack * dilation) for k in self._cnn_keys}
        new_spaces = dict(env.observation_space.spaces)
        for k in self._cnn_keys:
            sp = env.observation_space[k]
            shape = (num_stack * sp.shape[0],) + tuple(sp.shape[1:])
            new_spaces[k] = spaces.Box(
                np.repeat(sp.low, num_stack, axis=0), np.repeat(sp.high, num_stack, axis=0), shape, sp.dtype
            )
        self.observation_space = spaces.Dict(new_spaces)

    def _stacked(self, k: str) -> np.ndarray:
        frames = list(self._frames[k])[:: self.dilation][-self.num_stack :]
        return np.concatenate(frames, axis=0)

    def observation(self, obs):
        out = dict(obs)
        for k in self._cnn_keys:
            self._frames[k].append(obs[k])
            out[k] = self._stacked(k)
        return out

    def reset(self, *, seed=None, options=None):
        obs, info = self.env.reset(seed=seed, options=options)
        out = dict(obs)
        for k in self._cnn_keys:
            for _ in range(self.num_stack * self.dilation):
                self._frames[k].append(obs[k])
            out[k] = self._stacked(k)
        return out, info


class RewardAsObservation(ObservationWrapper):
    """Expose the last reward as an observation key
    (parity: sheeprl/envs/wrappers.py:185)."""

    def __init__(self, env: Env) -> None:
        super().__init__(env)
        new_spaces = dict(env.observation_space.spaces) if isinstance(env.observation_space, spaces.Dict) else {}
        new_spaces["reward"] = spaces.Box(-np.inf, np.inf, (1,), np.float32)
        self.observation_space = spaces.Dict(new_spaces)
        self._last_reward = 0.0

    def observation(self, obs):
        out = dict(obs)
        out["reward"] = np.array([self._last_reward], dtype=np.float32)
        return out

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        self._last_reward = float(r)
        return self.observation(obs), r, term, trunc, info

    def reset(self, *, seed=None, options=None):
        self._last_reward = 0.0
        return super().reset(seed=seed, options=options)


class ActionsAsObservation(ObservationWrapper):
    """Expose a (possibly dilated) stack of the last actions as an observation
    key (parity: sheeprl/envs/wrappers.py:258)."""

    def __init__(self, env: Env, num_stack: int, noop: Any, dilation: int = 1) -> None:
        super().__init__(env)
        if num_stack < 1:
            raise ValueError(f"num_stack must be >= 1, got {num_stack}")
        if dilation < 1:
            raise ValueError(f"dilation must be >= 1, got {dilation}")
        self.num_stack = num_stack
        self.dilation = dilation
        act = env.action_space
        if isinstance(act, spaces.Discrete):
            self._per_action = act.n
            if not isinstance(noop, int):
                raise ValueError("noop must be an int for discrete action spaces")
        elif isinstance(act, spaces.MultiDiscrete):
            self._per_action = int(act.nvec.sum())
            if not isinstance(noop, (list, tuple)):
                raise ValueError("noop must be a list for multi-discrete action spaces")
        elif isinstance(act, spaces.Box):
            self._per_action = int(np.prod(act.shape))
            if not isinstance(noop, float):
                raise ValueError("noop must be a float for continuous action spaces")
        else:
            raise ValueError(f"unsupported action space {act}")
        self.noop = noop
        self._actions: deque = deque(maxlen=num_stack * dilation)
        new_spaces = dict(env.observation_space.spaces)
        new_spaces["action_stack"] = spaces.Box(0.0 if not isinstance(act, spaces.Box) else -np.inf, np.inf,
                                                (num_stack * self._per_action,), np.float32)
        self.observation_space = spaces.Dict(new_spaces)

    def _encode(self, action: Any) -> np.ndarray:
        act = self.env.action_space
        if isinstance(act, spaces.Discrete):
            v = np.zeros(self._per_action, dtype=np.float32)
            v[int(action)] = 1.0
            return v
        if isinstance(act, spaces.MultiDiscrete):
            v = np.zeros(self._per_action, dtype=np.float32)
            off = 0
            for a, n in zip(np.asarray(action).reshape(-1), act.nvec):
                v[off + int(a)] = 1.0
                off += int(n)
            return v
        return np.asarray(action, dtype=np.float32).reshape(-1)

    def _stacked(self) -> np.ndarray:
        acts = list(self._actions)[:: self.dilation][-self.num_stack :]
        return np.concatenate(acts, axis=0)

    def observation(self, obs):
        out = dict(obs)
        out["action_stack"] = self._stacked()
        return out

    def step(self, action):
        self._actions.append(self._encode(action))
        obs, r, term, trunc, info = self.env.step(action)
        return self.observation(obs), r, term, trunc, info

    def reset(self, *, seed=None, options=None):
        obs, info = self.env.reset(seed=seed, options=options)
        noop = self.noop
        act = self.env.action_space
        if isinstance(act, spaces.Discrete):
            enc = self._encode(noop)
        elif isinstance(act, spaces.MultiDiscrete):
            enc = self._encode(np.asarray(noop))
        else:
            enc = np.full(self._per_action, float(noop), dtype=np.float32)
        for _ in range(self.num_stack * self.dilation):
            self._actions.append(enc)
        return self.observation(obs), info


class RestartOnException(Wrapper):
    """Rebuild a crashed env and signal the caller through
    ``info["restart_on_exception"]`` (parity: sheeprl/envs/wrappers.py:74-123;
    consumed by Dreamer-V3's buffer patch-up, dreamer_v3.py:595-608)."""

    def __init__(self, env_fn: Callable[[], Env], maxfails: int = 5, window: float = 60.0) -> None:
        self._env_fn = env_fn
        super().__init__(env_fn())
        self.maxfails = maxfails
        self.window = window
        self._fails = 0
        self._last_fail = 0.0

    def _rebuild(self) -> None:
        now = time.time()
        if now - self._last_fail > self.window:
            self._fails = 0
        self._fails += 1
        self._last_fail = now
        if self._fails > self.maxfails:
            raise RuntimeError(f"env failed more than {self.maxfails} times in {self.window}s")
        try:
            self.env.close()
        except Exception:
            pass
        self.env = self._env_fn()

    def step(self, action):
        try:
            return self.env.step(action)
        except Exception:
            self._rebuild()
            obs, info = self.env.reset()
            info = dict(info)
            info["restart_on_exception"] = True
            return obs, 0.0, False, True, info

    def reset(self, *, seed=None, options=None):
        try:
            return self.env.reset(seed=seed, options=options)
        except Exception:
            self._rebuild()
            obs, info = self.env.reset(seed=seed, options=options)
            info = dict(info)
            info["restart_on_exception"] = True
            return obs, info


class DictObservation(ObservationWrapper):
    """Wrap a flat Box observation into ``{"state": obs}``
    (the dict-ification step of sheeprl/utils/env.py:161-198)."""

    def __init__(self, env: Env, key: str = "state") -> None:
        super().__init__(env)
        self.key = key
        if isinstance(env.observation_space, spaces.Dict):
            self.observation_space = env.observation_space
            self._passthrough = True
        else:
            self.observation_space = spaces.Dict({key: env.observation_space})
            self._passthrough = False

    def observation(self, obs):
        return obs if self._passthrough else {self.key: obs}


class GrayscaleObservation(ObservationWrapper):
    """RGB [3,H,W] uint8 -> grayscale [1,H,W] uint8 for the given keys."""

    def __init__(self, env: Env, keys: list) -> None:
        super().__init__(env)
        self.keys = [k for k in keys if k in env.observation_space.spaces]
        new_spaces = dict(env.observation_space.spaces)
        for k in self.keys:
            sp = env.observation_space[k]
            new_spaces[k] = spaces.Box(0, 255, (1,) + tuple(sp.shape[1:]), np.uint8)
        self.observation_space = spaces.Dict(new_spaces)

    def observation(self, obs):
        out = dict(obs)
        for k in self.keys:
            img = obs[k].astype(np.float32)
            gray = 0.299 * img[0] + 0.587 * img[1] + 0.114 * img[2]
            out[k] = gray[None].astype(np.uint8)
        return out


def resize_area(img: np.ndarray, size: Tuple[int, int]) -> np.ndarray:
    """Simple area/nearest resize for [C,H,W] uint8 without OpenCV."""
    c, h, w = img.shape
    th, tw = size
    if (h, w) == (th, tw):
        return img
    if h % th == 0 and w % tw == 0:
        fh, fw = h // th, w // tw
        return img.reshape(c, th, fh, tw, fw).mean(axis=(2, 4)).astype(img.dtype)
    ys = (np.arange(th) * h / th).astype(np.int64)
    xs = (np.arange(tw) * w / tw).astype(np.int64)
    return img[:, ys][:, :, xs]


class ResizeObservation(ObservationWrapper):
    def __init__(self, env: Env, size: Tuple[int, int], keys: list) -> None:
        super().__init__(env)
        self.size = tuple(size)
        self.keys = [k for k in keys if k in env.observation_space.spaces]
        new_spaces = dict(env.observation_space.spaces)
        for k in self.keys:
            sp = env.observation_space[k]
            new_spaces[k] = spaces.Box(0, 255, (sp.shape[0],) + self.size, sp.dtype)
        self.observation_space = spaces.Dict(new_spaces)

    def observation(self, obs):
        out = dict(obs)
        for k in self.keys:
            out[k] = resize_area(obs[k], self.size)
        return out


class RecordVideo(Wrapper):
    """Record episode frames (parity role: gym's RecordVideoV0 used via
    cfg.env.capture_video).  No video codecs ship in this image, so episodes
    are saved as compressed ``.npz`` frame stacks under ``video_dir`` —
    loadable with ``np.load(...)["frames"]`` ([T,C,H,W] uint8)."""

    def __init__(self, env: Env, video_dir: str, key: str = "rgb", every_n_episodes: int = 1) -> None:
        super().__init__(env)
        import os

        self.video_dir = video_dir
        os.makedirs(video_dir, exist_ok=True)
        self.key = key
        self.every = max(1, every_n_episodes)
        self._frames: list = []
        self._episode = 0

    def _grab(self, obs: Any) -> None:
        if self._episode % self.every != 0:
            return
        frame = obs.get(self.key) if isinstance(obs, dict) else obs
        if frame is not None:
            self._frames.append(np.asarray(frame))

    def reset(self, *, seed=None, options=None):
        obs, info = self.env.reset(seed=seed, options=options)
        self._frames = []
        self._grab(obs)
        return obs, info

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        self._grab(obs)
        if term or trunc:
            if self._frames and self._episode % self.every == 0:
                import os

                path = os.path.join(self.video_dir, f"episode_{self._episode}.npz")
                np.savez_compressed(path, frames=np.stack(self._frames))
            self._episode += 1
            self._frames = []
        return obs, r, term, trunc, info
