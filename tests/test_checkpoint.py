import os

import torch

from torched_impala_amd import checkpoint
from torched_impala_amd.config import Flags
from torched_impala_amd.learner import build_optimizer
from torched_impala_amd.models import MLPPolicy


def test_save_load_bitwise(tmp_path):
    flags = Flags(env="cartpole", total_steps=1000)
    model = MLPPolicy((4,), 2)
    opt, sched = build_optimizer(flags, model)
    # run a step so optimizer state exists
    loss = model(dict(frame=torch.randn(1, 2, 4), reward=torch.zeros(1, 2),
                      done=torch.zeros(1, 2, dtype=torch.bool),
                      last_action=torch.zeros(1, 2, dtype=torch.long)),
                 ())[0]["policy_logits"].sum()
    loss.backward()
    opt.step()
    sched.step()
    path = str(tmp_path / "model.tar")
    checkpoint.save(path, model, opt, sched, step=640, flags=flags)
    assert os.path.exists(path)

    model2 = MLPPolicy((4,), 2)
    opt2, sched2 = build_optimizer(flags, model2)
    ckpt = checkpoint.load(path, model2, opt2, sched2)
    assert ckpt["step"] == 640
    assert ckpt["flags"]["env"] == "cartpole"
    for (k1, v1), (k2, v2) in zip(model.state_dict().items(),
                                  model2.state_dict().items()):
        assert k1 == k2
        assert torch.equal(v1, v2), f"param {k1} not bitwise equal"
    # optimizer square-avg state restored
    s1 = opt.state_dict()["state"]
    s2 = opt2.state_dict()["state"]
    for k in s1:
        assert torch.equal(s1[k]["square_avg"], s2[k]["square_avg"])


def test_frozen_dict_keys(tmp_path):
    """The checkpoint format contract: exact top-level keys."""
    flags = Flags()
    model = MLPPolicy((4,), 2)
    opt, sched = build_optimizer(flags, model)
    path = str(tmp_path / "m.tar")
    checkpoint.save(path, model, opt, sched, 0, flags)
    ckpt = torch.load(path, weights_only=False)
    assert set(ckpt.keys()) == {"model_state_dict", "optimizer_state_dict",
                                "scheduler_state_dict", "step", "flags"}
