"""Checkpoint format (FROZEN — the product contract).

A single ``torch.save`` dict:

    {"model_state_dict":     model.state_dict(),   # frozen param names
     "optimizer_state_dict": optimizer.state_dict(),
     "scheduler_state_dict": scheduler.state_dict(),
     "step":                 int,   # env (agent) frames consumed
     "flags":                vars(Flags)}

Parameter naming follows the torchbeast-lineage convention documented in
models/nets.py (conv1/conv2/conv3/fc/core/policy/baseline, ...). Since
the reference repo was unreadable, this definition is the
"same checkpoint format" contract and is covered by
tests/test_checkpoint.py (save -> load bitwise identity, resume step).
"""
from __future__ import annotations

import dataclasses
import logging
import os

import torch

log = logging.getLogger(__name__)


def save(path: str, model, optimizer, scheduler, step: int, flags):
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    tmp = path + ".tmp"
    # clone: FlatRMSprop params are views into one flat storage — cloning
    # keeps each checkpoint entry self-contained (and small)
    torch.save({
        "model_state_dict": {k: v.detach().clone()
                             for k, v in model.state_dict().items()},
        "optimizer_state_dict": optimizer.state_dict() if optimizer else None,
        "scheduler_state_dict": scheduler.state_dict() if scheduler else None,
        "step": int(step),
        "flags": dataclasses.asdict(flags) if dataclasses.is_dataclass(flags)
                 else dict(flags),
    }, tmp)
    os.replace(tmp, path)
    log.info("checkpointed step=%d -> %s", step, path)


def load(path: str, model=None, optimizer=None, scheduler=None,
         map_location="cpu"):
    ckpt = torch.load(path, map_location=map_location, weights_only=False)
    if model is not None:
        model.load_state_dict(ckpt["model_state_dict"])
    if optimizer is not None and ckpt.get("optimizer_state_dict"):
        optimizer.load_state_dict(ckpt["optimizer_state_dict"])
    if scheduler is not None and ckpt.get("scheduler_state_dict"):
        scheduler.load_state_dict(ckpt["scheduler_state_dict"])
    return ckpt
