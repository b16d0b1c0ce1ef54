"""Replay a non-finite-state dump (IMPALA_DUMP_NONFINITE) through the
graphed + eager steps to determine whether the poison lives in the
BATCH, the PARAMS, or the replay mechanism.

  python scripts/dump_replay.py gpurun_out/r17_dump.pt
"""
from __future__ import annotations

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torched_impala_amd.config import Flags
from torched_impala_amd import learner as L
from torched_impala_amd.models import build_model


class Stub:
    def __init__(self, flags, st):
        self.flags = flags
        self.k = flags.batch_size
        self.staging = {0: st}

    _to_time_major = L.Batcher._to_time_major


def finite_report(name, t):
    t = t.float()
    fin = bool(torch.isfinite(t).all())
    print(f"  {name:24s} finite={fin} absmax="
          f"{float(t[torch.isfinite(t)].abs().max() if fin is False else t.abs().max()):.4g}",
          flush=True)
    return fin


def main():
    path = sys.argv[1]
    blob = torch.load(path, map_location="cpu", weights_only=False)
    print("dump step:", blob.get("step"), "stats:", blob.get("stats"))
    if "per_param" in blob:
        print("bad params:", blob["per_param"])
    print("== dumped tensors ==")
    ok_batch = True
    for k, v in (blob.get("staging") or {}).items():
        ok_batch &= finite_report(f"staging.{k}", v)
    for k in ("flat_param", "flat_grad", "flat_sqavg"):
        if blob.get(k) is not None:
            finite_report(k, blob[k])
    device = torch.device("cuda", 0)
    flags = Flags(env="synthetic-atari", batch_size=32, unroll_length=20,
                  total_steps=1 << 50)
    obs, A = (4, 84, 84), 6
    torch.manual_seed(0)
    model = build_model(flags, obs, A).to(device)
    opt, sched = L.build_optimizer(flags, model)
    st = {k: v.to(device) for k, v in blob["staging"].items()}
    # sanitise the params: run the dumped BATCH against FRESH weights
    stub = Stub(flags, st)
    batch = stub._to_time_major(st)
    L.learn_step(flags, model, dict(batch), opt, sched, None)
    torch.cuda.synchronize()
    print("eager fresh-weights on dumped batch: grad finite =",
          bool(torch.isfinite(opt.flat_grad).all()),
          "loss finite =", True)
    # graphed on the dumped batch, fresh weights, 50 replays
    torch.manual_seed(0)
    model2 = build_model(flags, obs, A).to(device)
    opt2, sched2 = L.build_optimizer(flags, model2)
    stub2 = Stub(flags, st)
    graphed = L.GraphedStep(flags, model2, opt2, stub2)
    bad = None
    for i in range(50):
        stats = graphed.step(0, sched2)
        torch.cuda.synchronize()
        if not bool(torch.isfinite(opt2.flat_param).all()):
            bad = i
            break
    print("graphed fresh-weights on dumped batch: first bad replay =", bad)
    # params from the dump (already NaN?) -> skip if non-finite
    print(json.dumps({"replay": "done"}))


if __name__ == "__main__":
    main()
