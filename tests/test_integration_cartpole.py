"""Plumbing integration (BASELINE.json:7 config #1):
CartPole MLP, CPU, real actor processes + ring transport + learner.
Asserts: no deadlock, finite losses, checkpointing, and (longer budget,
separate test) mean return trends up."""
import os

import pytest
import torch

from torched_impala_amd.config import Flags
from torched_impala_amd import learner


@pytest.mark.timeout(300)
def test_cartpole_train_smoke(tmp_path):
    flags = Flags(env="cartpole", num_actors=2, batch_size=4,
                  unroll_length=10, total_steps=100000,
                  device="cpu", xpid="smoke", savedir=str(tmp_path),
                  metrics_jsonl=str(tmp_path / "metrics.jsonl"),
                  seed=7)
    stats = learner.train(flags, max_learner_steps=12)
    assert stats["learner_steps"] == 12
    assert stats["step_frames"] == 12 * 4 * 10
    for k in ("total_loss", "pg_loss", "baseline_loss", "entropy_loss"):
        assert k in stats and stats[k] == stats[k], f"{k} is NaN"
    ckpt = os.path.join(str(tmp_path), "smoke", "model.tar")
    assert os.path.exists(ckpt)
    assert os.path.exists(str(tmp_path / "metrics.jsonl"))


@pytest.mark.timeout(300)
@pytest.mark.slow
def test_cartpole_learns(tmp_path):
    """Return trends up over a modest budget (seeded; generous threshold)."""
    flags = Flags(env="cartpole", num_actors=2, batch_size=8,
                  unroll_length=20, total_steps=120000,
                  learning_rate=0.002, entropy_cost=0.01,
                  device="cpu", disable_checkpoint=True, seed=1)
    returns = []

    stats = learner.train(flags, max_learner_steps=400)
    # untrained CartPole random policy averages ~20; trained should exceed 50
    assert stats["mean_return"] == stats["mean_return"], "no episodes finished"
    assert stats["mean_return"] > 45, f"no learning: {stats['mean_return']}"


@pytest.mark.timeout(300)
def test_cartpole_test_mode(tmp_path):
    flags = Flags(env="cartpole", device="cpu")
    out = learner.test(flags, num_episodes=2)
    assert out["mean_return"] >= 1
