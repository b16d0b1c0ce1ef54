"""Failure detection: dead actors are respawned and
training continues; checkpoint resume continues the step count."""
import os

import pytest
import torch

from torched_impala_amd.config import Flags
from torched_impala_amd import learner


@pytest.mark.timeout(300)
def test_actor_respawn_keeps_training(tmp_path, monkeypatch):
    """Kill one actor mid-run; the learner must respawn it and finish."""
    flags = Flags(env="cartpole", num_actors=2, batch_size=4,
                  unroll_length=10, total_steps=1 << 40,
                  device="cpu", disable_checkpoint=True, seed=11)
    killed = {"done": False}
    procs_ref = {}

    orig_start = learner.start_actors

    def capture_start(*args, **kw):
        procs = orig_start(*args, **kw)
        procs_ref["procs"] = procs
        return procs

    monkeypatch.setattr(learner, "start_actors", capture_start)

    def on_step(steps, frames, stats):
        if steps == 3 and not killed["done"]:
            procs_ref["procs"][0].terminate()
            killed["done"] = True

    stats = learner.train(flags, max_learner_steps=15, on_step=on_step)
    assert stats["learner_steps"] == 15
    assert killed["done"]


@pytest.mark.timeout(300)
def test_checkpoint_resume_continues_steps(tmp_path):
    common = dict(env="cartpole", num_actors=1, batch_size=2,
                  unroll_length=5, total_steps=1 << 40, device="cpu",
                  savedir=str(tmp_path), xpid="resume", seed=5)
    flags = Flags(**common)
    stats1 = learner.train(flags, max_learner_steps=4)
    frames1 = stats1["step_frames"]
    assert os.path.exists(str(tmp_path / "resume" / "model.tar"))

    stats2 = learner.train(Flags(**common), max_learner_steps=3)
    # resumed run starts counting from the checkpointed frame count
    assert stats2["step_frames"] == frames1 + 3 * 2 * 5
