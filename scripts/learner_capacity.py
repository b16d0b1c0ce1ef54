"""Diagnostic (NOT the contract bench): pure learner-step capacity.

Pre-stages one synthetic device batch and runs learn_step in a loop —
measures the GPU-side ceiling (forward/V-trace/loss/backward/optimizer)
with the actor/transport pipeline taken out. Use to target kernel work;
the headline number remains bench.py (whole system).

  python scripts/learner_capacity.py [--use-lstm] [--model shallow|deep]
      [--batch-size 32] [--unroll-length 20] [--steps 50]
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from torched_impala_amd.config import Flags
from torched_impala_amd.learner import build_optimizer, learn_step
from torched_impala_amd.models import build_model


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="shallow", choices=["shallow", "deep", "mlp"])
    p.add_argument("--use-lstm", action="store_true")
    p.add_argument("--batch-size", type=int, default=32)
    p.add_argument("--unroll-length", type=int, default=20)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--no-hip-kernels", action="store_true")
    p.add_argument("--num-actions", type=int, default=6)
    p.add_argument("--compute-dtype", default="fp32", choices=["fp32", "bf16"])
    p.add_argument("--channels-last", action="store_true")
    p.add_argument("--graphs", action="store_true",
                   help="hipGraph-captured step (the product path) "
                        "instead of eager learn_step")
    args = p.parse_args()

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    obs = (3, 72, 96) if args.model == "deep" else (4, 84, 84)
    A = 15 if args.model == "deep" else args.num_actions
    flags = Flags(env="synthetic-atari", model=args.model,
                  use_lstm=args.use_lstm,
                  batch_size=args.batch_size, unroll_length=args.unroll_length,
                  total_steps=1 << 50,
                  use_hip_kernels=not args.no_hip_kernels,
                  compute_dtype=args.compute_dtype)
    torch.manual_seed(0)
    model = build_model(flags, obs, A).to(device)
    if args.channels_last:
        model = model.to(memory_format=torch.channels_last)
    model.force_eager_core = not flags.use_hip_kernels
    optimizer, scheduler = build_optimizer(flags, model)

    T, B = args.unroll_length, args.batch_size
    batch = dict(
        frame=torch.randint(0, 256, (T + 1, B, *obs), dtype=torch.uint8,
                            device=device),
        reward=torch.randn(T + 1, B, device=device),
        done=torch.rand(T + 1, B, device=device) < 0.01,
        action=torch.randint(0, A, (T + 1, B), device=device),
        last_action=torch.randint(0, A, (T + 1, B), device=device),
        behavior_logits=torch.randn(T + 1, B, A, device=device),
        episode_return=torch.zeros(T + 1, B, device=device),
        episode_step=torch.zeros(T + 1, B, dtype=torch.int32, device=device),
    )
    if args.use_lstm:
        batch["initial_core_state"] = torch.zeros(2, B, 256, device=device)

    if args.graphs and device.type == "cuda":
        from torched_impala_amd.learner import GraphedStep

        st = {}
        for kk, v in batch.items():
            if kk == "initial_core_state":       # (2,B,H) -> (B,2,1,H)
                st[kk] = v.permute(1, 0, 2).unsqueeze(2).contiguous()
            else:                                # (T+1,B,...) -> (B,T+1,1,...)
                st[kk] = v.transpose(0, 1).unsqueeze(2).contiguous()

        class _Stub:
            def __init__(self, flags, st):
                self.flags = flags
                self.k = flags.batch_size
                self.staging = {0: st}
            from torched_impala_amd.learner import Batcher as _B
            _to_time_major = _B._to_time_major
            del _B

        graphed = GraphedStep(flags, model, optimizer, _Stub(flags, st))

        def one_step():
            graphed.step(0, scheduler)
    else:
        def one_step():
            learn_step(flags, model, dict(batch), optimizer, scheduler, None)

    for _ in range(args.warmup):
        one_step()
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    ms = dt / args.steps * 1000
    fps = args.steps * T * B / dt
    mem_gb = (torch.cuda.max_memory_allocated() / 2**30
              if device.type == "cuda" else 0.0)
    print(json.dumps(dict(
        diagnostic="learner_capacity", model=args.model, lstm=args.use_lstm,
        hip=not args.no_hip_kernels, dtype=args.compute_dtype,
        channels_last=args.channels_last, graphs=bool(args.graphs), T=T, B=B,
        ms_per_step=round(ms, 3), frames_per_sec=round(fps, 1),
        peak_hbm_gib=round(mem_gb, 2))))


if __name__ == "__main__":
    main()
