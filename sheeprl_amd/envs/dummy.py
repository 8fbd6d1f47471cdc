"""Deterministic fake environments used by the test-suite.

Parity with sheeprl/envs/dummy.py: Continuous / Discrete /
MultiDiscrete dummies with dict observations {"rgb": [3,64,64] uint8,
"state": [4] float32} whose pixel values follow a deterministic counter so
tests are reproducible.  Image convention in this framework is channel-first.
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np

from sheeprl_amd.envs import spaces
from sheeprl_amd.envs.core import Env


class _DummyBase(Env):
    def __init__(self, image_size: Tuple[int, int, int] = (3, 64, 64), n_steps: int = 128, state_dim: int = 4) -> None:
        self.image_size = image_size
        self.n_steps = n_steps
        self.state_dim = state_dim
        self.observation_space = spaces.Dict(
            {
                "rgb": spaces.Box(0, 255, image_size, np.uint8),
                "state": spaces.Box(-np.inf, np.inf, (state_dim,), np.float32),
            }
        )
        self.reward_range = (0.0, 1.0)
        self._t = 0

    def _obs(self) -> dict:
        return {
            "rgb": np.full(self.image_size, self._t % 256, dtype=np.uint8),
            "state": np.full((self.state_dim,), self._t, dtype=np.float32),
        }

    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None):
        self._t = 0
        return self._obs(), {}

    def _step_common(self):
        self._t += 1
        done = self._t >= self.n_steps
        return self._obs(), 1.0, done, False, {}


class ContinuousDummyEnv(_DummyBase):
    def __init__(self, action_dim: int = 2, **kw) -> None:
        super().__init__(**kw)
        self.action_space = spaces.Box(-1.0, 1.0, (action_dim,), np.float32)

    def step(self, action):
        return self._step_common()


class DiscreteDummyEnv(_DummyBase):
    def __init__(self, action_dim: int = 4, **kw) -> None:
        super().__init__(**kw)
        self.action_space = spaces.Discrete(action_dim)

    def step(self, action):
        return self._step_common()


class MultiDiscreteDummyEnv(_DummyBase):
    def __init__(self, action_dims: Tuple[int, ...] = (2, 2), **kw) -> None:
        super().__init__(**kw)
        self.action_space = spaces.MultiDiscrete(action_dims)

    def step(self, action):
        return self._step_common()
