"""Environments.

No gym/ale dependency exists in this image, so the engine
ships an internal CartPole-v1 (public dynamics) and synthetic-frame
environments matching the benchmark observation shapes
(BASELINE.json:8,11: Atari-shaped 84x84x4 uint8, DMLab-shaped 72x96x3).
"""
from __future__ import annotations

from torched_impala_amd.envs.cartpole import CartPoleEnv
from torched_impala_amd.envs.synthetic import SyntheticFrameEnv


def make_env(flags, seed: int = 0):
    """Environment factory (the reference's make_env analog)."""
    name = flags.env
    if name == "cartpole":
        return CartPoleEnv(seed=seed)
    if name == "synthetic-atari":
        hw = flags.synthetic_frame_hw or "84x84x4"
        h, w, c = (int(x) for x in hw.split("x"))
        return SyntheticFrameEnv(
            height=h, width=w, channels=c,
            num_actions=flags.synthetic_num_actions,
            episode_len=flags.synthetic_episode_len, seed=seed)
    if name == "synthetic-dmlab":
        hw = flags.synthetic_frame_hw or "72x96x3"
        h, w, c = (int(x) for x in hw.split("x"))
        return SyntheticFrameEnv(
            height=h, width=w, channels=c,
            num_actions=flags.synthetic_num_actions,
            episode_len=flags.synthetic_episode_len, seed=seed)
    raise ValueError(f"unknown env {name!r}")


def env_spec(flags):
    """(obs_shape CHW or flat, obs_dtype, num_actions) for model construction."""
    env = make_env(flags, seed=0)
    try:
        return env.obs_shape, env.obs_dtype, env.num_actions
    finally:
        env.close()


__all__ = ["CartPoleEnv", "SyntheticFrameEnv", "make_env", "env_spec"]
