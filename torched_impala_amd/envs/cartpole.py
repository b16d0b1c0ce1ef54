"""Internal CartPole-v1 (classic control; public dynamics, no gym).

Matches the OpenAI Gym CartPole-v1 dynamics and termination so that the
plumbing config (BASELINE.json:7: "CartPole-v1 MLP policy, single-process
actor+learner on CPU") is runnable with zero external dependencies.
"""
from __future__ import annotations

import math

import numpy as np


class CartPoleEnv:
    """reset() -> obs; step(a) -> (obs, reward, done, info).

    Observation: float32[4] (x, x_dot, theta, theta_dot).
    Actions: {0, 1}. Reward +1 per step; episode caps at 500 steps.
    """

    GRAVITY = 9.8
    MASSCART = 1.0
    MASSPOLE = 0.1
    TOTAL_MASS = MASSCART + MASSPOLE
    LENGTH = 0.5  # half pole length
    POLEMASS_LENGTH = MASSPOLE * LENGTH
    FORCE_MAG = 10.0
    TAU = 0.02
    THETA_THRESHOLD = 12 * 2 * math.pi / 360
    X_THRESHOLD = 2.4
    MAX_STEPS = 500

    obs_dtype = np.float32
    num_actions = 2

    def __init__(self, seed: int = 0):
        self.rng = np.random.default_rng(seed)
        self.state = None
        self.steps = 0
        self.obs_shape = (4,)

    def reset(self) -> np.ndarray:
        self.state = self.rng.uniform(-0.05, 0.05, size=4).astype(np.float64)
        self.steps = 0
        return self.state.astype(np.float32)

    def step(self, action: int):
        assert self.state is not None, "call reset() first"
        x, x_dot, theta, theta_dot = self.state
        force = self.FORCE_MAG if action == 1 else -self.FORCE_MAG
        costheta, sintheta = math.cos(theta), math.sin(theta)
        temp = (force + self.POLEMASS_LENGTH * theta_dot**2 * sintheta) / self.TOTAL_MASS
        thetaacc = (self.GRAVITY * sintheta - costheta * temp) / (
            self.LENGTH * (4.0 / 3.0 - self.MASSPOLE * costheta**2 / self.TOTAL_MASS))
        xacc = temp - self.POLEMASS_LENGTH * thetaacc * costheta / self.TOTAL_MASS
        x = x + self.TAU * x_dot
        x_dot = x_dot + self.TAU * xacc
        theta = theta + self.TAU * theta_dot
        theta_dot = theta_dot + self.TAU * thetaacc
        self.state = np.array([x, x_dot, theta, theta_dot], dtype=np.float64)
        self.steps += 1
        done = bool(
            x < -self.X_THRESHOLD or x > self.X_THRESHOLD
            or theta < -self.THETA_THRESHOLD or theta > self.THETA_THRESHOLD
            or self.steps >= self.MAX_STEPS)
        return self.state.astype(np.float32), 1.0, done, {}

    def close(self):
        pass
