"""Synthetic-frame environment (the benchmark's data source).

BASELINE.json:5 fixes the benchmark on "synthetic frames / random-init
weights": deterministic pseudo-random uint8 frames of the Atari shape
(4, 84, 84) or the DMLab shape (3, 72, 96), CHW, with a fixed per-env
seed so multi-process tests are reproducible.

Frame generation is deliberately cheap (cached bank of frames, indexed by
step counter) so the benchmark measures the engine, not numpy's RNG.
"""
from __future__ import annotations

import numpy as np

# one frame bank per (shape) per PROCESS: the bank is seeded identically
# for every env, so per-instance copies were pure waste — 16 envs/actor
# x 1.8 MiB duplicated banks blew the actor working set past the EPYC's
# 32 MiB/CCD L3 and capped the node's aggregate actor throughput
# (profiles/r4_mfma_attribution.md supply curve).
_BANK_CACHE: dict = {}


def _shared_bank(shape):
    b = _BANK_CACHE.get(shape)
    if b is None:
        rng = np.random.default_rng(12345)
        b = rng.integers(0, 256, size=(SyntheticFrameEnv.FRAME_BANK, *shape),
                         dtype=np.uint8)
        _BANK_CACHE[shape] = b
    return b


class SyntheticFrameEnv:
    """Atari/DMLab-shaped synthetic environment.

    Observation: uint8[C, H, W] (CHW, as the models consume it).
    Reward: deterministic pseudo-random in {-1, 0, +1}.
    Episode ends every `episode_len` steps.
    """

    obs_dtype = np.uint8

    FRAME_BANK = 64  # distinct frames cycled by (step + seed) — cheap but non-constant

    def __init__(self, height=84, width=84, channels=4, num_actions=6,
                 episode_len=1000, seed=0):
        self.obs_shape = (channels, height, width)
        self.num_actions = num_actions
        self.episode_len = episode_len
        self.seed = seed
        self.bank = _shared_bank(self.obs_shape)  # per-process, shared
        rng2 = np.random.default_rng(seed)
        self.reward_bank = (rng2.integers(0, 3, size=1024).astype(np.float32) - 1.0)
        # stagger the first episode boundary per env (same-seed envs stay
        # identical) so batched envs don't all emit done on the same step
        self.t = int(seed) % episode_len
        self.global_t = int(seed) * 977

    def _frame(self) -> np.ndarray:
        return self.bank[self.global_t % self.FRAME_BANK]

    def reset(self) -> np.ndarray:
        self.t = 0
        self.global_t += 1
        return self._frame()

    def step(self, action: int):
        self.t += 1
        self.global_t += 1
        reward = float(self.reward_bank[(self.global_t + int(action)) % 1024])
        done = self.t >= self.episode_len
        return self._frame(), reward, done, {}

    def close(self):
        pass

