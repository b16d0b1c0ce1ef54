"""Diagnostic: aggregate actor throughput on THIS machine.

Spawns `--actors` processes each running the real Rollout loop (model
forward + env stepping, no ring buffer) for `--seconds`, reports
frames/sec per actor and aggregate. Use to size envs_per_actor and to
attribute bench variance to the actor side.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _worker(args, idx, q, barrier=None):
    import torch
    if args.pin:
        try:
            ncpu = os.cpu_count() or 1
            os.sched_setaffinity(0, {idx % ncpu})
        except OSError:
            pass
    torch.set_num_threads(max(1, args.threads))
    from torched_impala_amd.config import Flags
    from torched_impala_amd.models import build_model
    from torched_impala_amd.actor import Rollout
    flags = Flags(env="synthetic-atari", envs_per_actor=args.envs,
                  use_lstm=args.use_lstm, seed=idx, actor_jit=args.jit)
    model = build_model(flags, (4, 84, 84), 6)
    model.eval()
    roll = Rollout(flags, idx, model)
    for _ in range(5):
        roll._advance()
    if barrier is not None:
        barrier.wait(timeout=300)  # synchronized windows: honest overlap
    n = 0
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < args.seconds:
        roll._advance()
        n += args.envs
    q.put(n / (time.perf_counter() - t0))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--actors", type=int, default=32)
    p.add_argument("--envs", type=int, default=16)
    p.add_argument("--seconds", type=float, default=8.0)
    p.add_argument("--use-lstm", action="store_true")
    p.add_argument("--threads", type=int, default=1)
    p.add_argument("--jit", default="native-bf16",
                   choices=["off", "safe", "frozen", "native", "native-bf16"])
    p.add_argument("--pin", action="store_true",
                   help="pin worker i to cpu i (distinct physical cores)")
    p.add_argument("--no-barrier", action="store_true")
    args = p.parse_args()
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    barrier = None if args.no_barrier else ctx.Barrier(args.actors)
    procs = [ctx.Process(target=_worker, args=(args, i, q, barrier))
             for i in range(args.actors)]
    for pr in procs:
        pr.start()
    rates = [q.get(timeout=180) for _ in range(args.actors)]
    for pr in procs:
        pr.join(timeout=30)
    agg = sum(rates)
    print(json.dumps(dict(
        diagnostic="actor_probe", actors=args.actors, envs=args.envs,
        threads=args.threads, jit=args.jit,
        lstm=args.use_lstm, per_actor_fps=round(agg / args.actors, 1),
        aggregate_fps=round(agg, 1))))


if __name__ == "__main__":
    main()
