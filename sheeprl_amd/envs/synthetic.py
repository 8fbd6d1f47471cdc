"""Synthetic pixel environments for offline benchmarking.

There is no network access (no ALE/Atari ROMs), so the Dreamer-V3 Atari-100K
benchmark runs against this synthetic stand-in: same observation
shape ([C,64,64] uint8), same discrete action arity as MsPacman (9), cheap
deterministic frame generation so the env never bottlenecks the GPU
measurement (bench.py declares ``data: synthetic``).
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np

from sheeprl_amd.envs import spaces
from sheeprl_amd.envs.core import Env


class SyntheticAtariEnv(Env):
    """Cheap deterministic 'Atari-shaped' env: scrolling uint8 pattern frames,
    sparse pseudo-random rewards, geometric episode termination."""

    def __init__(
        self,
        n_actions: int = 9,
        image_size: Tuple[int, int, int] = (3, 64, 64),
        episode_len: int = 1000,
        seed: int = 0,
    ) -> None:
        self.observation_space = spaces.Dict({"rgb": spaces.Box(0, 255, image_size, np.uint8)})
        self.action_space = spaces.Discrete(n_actions)
        self.image_size = image_size
        self.episode_len = episode_len
        c, h, w = image_size
        rng = np.random.default_rng(seed)
        # A bank of pre-generated noise rows; frames are cheap rolls of it.
        self._base = rng.integers(0, 256, size=(c, h, w), dtype=np.uint8)
        self._t = 0
        self._rng = rng

    def _obs(self) -> dict:
        frame = np.roll(self._base, shift=self._t % 64, axis=1)
        return {"rgb": frame}

    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None):
        if seed is not None:
            self._rng = np.random.default_rng(seed)
        self._t = 0
        return self._obs(), {}

    def step(self, action):
        self._t += 1
        reward = float(self._rng.random() < 0.05)
        terminated = bool(self._t >= self.episode_len)
        return self._obs(), reward, terminated, False, {}
