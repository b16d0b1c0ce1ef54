"""Reproducer hunt for the round-1 flaky async hipErrorInvalidValue
(VERDICT r1 item 5): many hipGraph captures sharing one process, with
H2D side-stream copies interleaved — the observed failure mode was an
invalid-argument error surfacing at a LATER copy_ after several
graph-capture train tests ran in-process.

Modes:
  python scripts/graph_flake_repro.py --iters 12            # default: mimic the suite (no explicit teardown)
  python scripts/graph_flake_repro.py --iters 12 --teardown # with GraphedStep.destroy() between iterations
  AMD_SERIALIZE_KERNEL=3 ... for the serialized bisect
Exit code 0 = no fault; the first HIP error aborts with a traceback.
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torched_impala_amd.config import Flags
from torched_impala_amd import learner as L
from torched_impala_amd.models import build_model


def staging(flags, obs, A, device, seed):
    torch.manual_seed(seed)
    T = flags.unroll_length
    k, n = flags.batch_size, 1
    st = dict(
        frame=torch.randint(0, 256, (k, T + 1, n, *obs), dtype=torch.uint8,
                            device=device),
        reward=torch.randn(k, T + 1, n, device=device),
        done=torch.rand(k, T + 1, n, device=device) < 0.05,
        action=torch.randint(0, A, (k, T + 1, n), device=device),
        last_action=torch.randint(0, A, (k, T + 1, n), device=device),
        behavior_logits=torch.randn(k, T + 1, n, A, device=device),
        episode_return=torch.zeros(k, T + 1, n, device=device),
        episode_step=torch.zeros(k, T + 1, n, dtype=torch.int32,
                                 device=device),
    )
    if flags.use_lstm:
        st["initial_core_state"] = torch.zeros(k, 2, n, 256, device=device)
    return st


class Stub:
    def __init__(self, flags, st0, st1):
        self.flags = flags
        self.k = flags.batch_size // flags.envs_per_actor
        self.staging = {0: st0, 1: st1}

    _to_time_major = L.Batcher._to_time_major


def one_iteration(i, args, device):
    use_lstm = bool(i % 2) if args.mix_lstm else args.use_lstm
    flags = Flags(env="synthetic-atari", batch_size=8, unroll_length=10,
                  use_lstm=use_lstm, total_steps=1 << 50)
    obs, A = (4, 84, 84), 6
    torch.manual_seed(1000 + i)
    model = build_model(flags, obs, A).to(device)
    opt, sched = L.build_optimizer(flags, model)
    st0 = staging(flags, obs, A, device, seed=2 * i)
    st1 = staging(flags, obs, A, device, seed=2 * i + 1)
    stub = Stub(flags, st0, st1)
    g = L.GraphedStep(flags, model, opt, stub)
    for j in (0, 1, 0, 1, 0, 1):
        g.step(j, sched)
    # interleave pinned H2D copies on a side stream (the Batcher shape)
    side = torch.cuda.Stream()
    host = torch.empty(4 << 20, dtype=torch.uint8, pin_memory=True)
    dev = torch.empty(4 << 20, dtype=torch.uint8, device=device)
    with torch.cuda.stream(side):
        for _ in range(8):
            dev.copy_(host, non_blocking=True)
            host.copy_(dev, non_blocking=True)
    torch.cuda.current_stream().wait_stream(side)
    if args.teardown:
        g.destroy()
    # else: drop the references and let GC do whatever it does (the
    # suite's original behavior)
    return float(g.outputs[0][0].item()) if args.teardown is False else 0.0


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=12)
    p.add_argument("--teardown", action="store_true")
    p.add_argument("--use-lstm", action="store_true")
    p.add_argument("--mix-lstm", action="store_true", default=True)
    args = p.parse_args()
    device = torch.device("cuda", 0)
    for i in range(args.iters):
        one_iteration(i, args, device)
        # a synchronizing API call is where the async fault surfaces
        torch.cuda.synchronize()
        print(f"iter {i} ok", flush=True)
    print("REPRO_CLEAN")


if __name__ == "__main__":
    main()
