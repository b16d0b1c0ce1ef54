"""End-to-end GPU training (real actor processes -> shm ring -> HIP
learner) on a MI355X box — marked gpu."""
import json
import subprocess
import sys
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.timeout(420)
def test_synthetic_atari_gpu_train():
    from torched_impala_amd.config import Flags
    from torched_impala_amd import learner
    flags = Flags(env="synthetic-atari", num_actors=4, batch_size=8,
                  unroll_length=20, total_steps=1 << 40,
                  disable_checkpoint=True, device="cuda", seed=3)
    stats = learner.train(flags, max_learner_steps=10)
    assert stats["learner_steps"] == 10
    for k in ("total_loss", "pg_loss", "baseline_loss", "entropy_loss"):
        assert stats[k] == stats[k], f"{k} is NaN"


@pytest.mark.timeout(420)
def test_synthetic_atari_gpu_train_lstm():
    from torched_impala_amd.config import Flags
    from torched_impala_amd import learner
    flags = Flags(env="synthetic-atari", num_actors=4, batch_size=8,
                  unroll_length=20, use_lstm=True, total_steps=1 << 40,
                  disable_checkpoint=True, device="cuda", seed=4)
    stats = learner.train(flags, max_learner_steps=8)
    assert stats["learner_steps"] == 8
    assert stats["total_loss"] == stats["total_loss"]


@pytest.mark.timeout(420)
def test_smoke_entry():
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import __graft_entry__
    __graft_entry__.smoke()
