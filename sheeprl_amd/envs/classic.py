"""Native classic-control environments (gymnasium is not in this image).

Physics match the standard CartPole-v1 and Pendulum-v1 dynamics so wall-clock
benchmarks are comparable with the reference's gym-based runs
(BASELINE.md PPO/A2C rows use a CartPole-like env).
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from sheeprl_amd.envs import spaces
from sheeprl_amd.envs.core import Env


class CartPoleEnv(Env):
    """Standard CartPole-v1 dynamics (Barto-Sutton-Anderson)."""

    def __init__(self) -> None:
        self.gravity = 9.8
        self.masscart = 1.0
        self.masspole = 0.1
        self.total_mass = self.masspole + self.masscart
        self.length = 0.5
        self.polemass_length = self.masspole * self.length
        self.force_mag = 10.0
        self.tau = 0.02
        self.theta_threshold = 12 * 2 * np.pi / 360
        self.x_threshold = 2.4
        high = np.array([self.x_threshold * 2, np.inf, self.theta_threshold * 2, np.inf], dtype=np.float32)
        self.observation_space = spaces.Box(-high, high, (4,), np.float32)
        self.action_space = spaces.Discrete(2)
        self._rng = np.random.default_rng()
        self.state = np.zeros(4, dtype=np.float64)

    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None):
        if seed is not None:
            self._rng = np.random.default_rng(seed)
        self.state = self._rng.uniform(-0.05, 0.05, size=4)
        return self.state.astype(np.float32), {}

    def step(self, action):
        x, x_dot, theta, theta_dot = self.state
        force = self.force_mag if int(action) == 1 else -self.force_mag
        costheta, sintheta = np.cos(theta), np.sin(theta)
        temp = (force + self.polemass_length * theta_dot**2 * sintheta) / self.total_mass
        thetaacc = (self.gravity * sintheta - costheta * temp) / (
            self.length * (4.0 / 3.0 - self.masspole * costheta**2 / self.total_mass)
        )
        xacc = temp - self.polemass_length * thetaacc * costheta / self.total_mass
        x = x + self.tau * x_dot
        x_dot = x_dot + self.tau * xacc
        theta = theta + self.tau * theta_dot
        theta_dot = theta_dot + self.tau * thetaacc
        self.state = np.array([x, x_dot, theta, theta_dot])
        terminated = bool(
            x < -self.x_threshold
            or x > self.x_threshold
            or theta < -self.theta_threshold
            or theta > self.theta_threshold
        )
        return self.state.astype(np.float32), 1.0, terminated, False, {}


class PendulumEnv(Env):
    """Standard Pendulum-v1 dynamics (continuous torque control)."""

    def __init__(self) -> None:
        self.max_speed = 8.0
        self.max_torque = 2.0
        self.dt = 0.05
        self.g = 10.0
        self.m = 1.0
        self.l = 1.0
        high = np.array([1.0, 1.0, self.max_speed], dtype=np.float32)
        self.observation_space = spaces.Box(-high, high, (3,), np.float32)
        self.action_space = spaces.Box(-self.max_torque, self.max_torque, (1,), np.float32)
        self._rng = np.random.default_rng()
        self.state = np.zeros(2, dtype=np.float64)

    def _obs(self) -> np.ndarray:
        th, thdot = self.state
        return np.array([np.cos(th), np.sin(th), thdot], dtype=np.float32)

    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None):
        if seed is not None:
            self._rng = np.random.default_rng(seed)
        self.state = self._rng.uniform([-np.pi, -1.0], [np.pi, 1.0])
        return self._obs(), {}

    def step(self, action):
        th, thdot = self.state
        u = float(np.clip(np.asarray(action).reshape(-1)[0], -self.max_torque, self.max_torque))
        angle = ((th + np.pi) % (2 * np.pi)) - np.pi
        cost = angle**2 + 0.1 * thdot**2 + 0.001 * u**2
        newthdot = thdot + (3 * self.g / (2 * self.l) * np.sin(th) + 3.0 / (self.m * self.l**2) * u) * self.dt
        newthdot = float(np.clip(newthdot, -self.max_speed, self.max_speed))
        newth = th + newthdot * self.dt
        self.state = np.array([newth, newthdot])
        return self._obs(), -float(cost), False, False, {}
