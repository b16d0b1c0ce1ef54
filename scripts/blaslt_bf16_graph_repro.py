"""Minimal standalone reproducer attempt for the hipBLASLt bf16-GEMM-
under-hipGraph gradient corruption (profiles/r2_nan_rootcause.md).

In the engine the corruption needed the FULL train loop (actors + H2D
copy stream + async publish) running at speed; this harness replays a
captured bf16 mm backward in a loop with a side-stream H2D copy churn
and checks outputs bitwise against a reference replay. If it trips, it
is an upstream-reportable testcase; if not, it documents that the
minimal form does not reproduce (the engine-level A/B remains the
evidence).

  python scripts/blaslt_bf16_graph_repro.py --iters 20000
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=20000)
    p.add_argument("--m", type=int, default=640)
    p.add_argument("--n", type=int, default=256)
    p.add_argument("--k", type=int, default=1568)
    args = p.parse_args()
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    x = torch.randn(args.m, args.k, device=dev, dtype=torch.bfloat16)
    w = torch.randn(args.n, args.k, device=dev, requires_grad=True)
    host = torch.empty(8 << 20, dtype=torch.uint8, pin_memory=True)
    devbuf = torch.empty(8 << 20, dtype=torch.uint8, device=dev)
    side = torch.cuda.Stream()

    def step():
        wb = w.to(torch.bfloat16)
        y = torch.nn.functional.linear(x, wb)
        loss = (y.float() ** 2).sum()
        w.grad = None
        loss.backward()
        return w.grad

    # capture
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            step()
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        grad = step()
    g.replay()
    torch.cuda.synchronize()
    ref = grad.clone()

    bad = 0
    for i in range(args.iters):
        with torch.cuda.stream(side):  # churn: async H2D/D2H like Batcher
            devbuf.copy_(host, non_blocking=True)
            host.copy_(devbuf, non_blocking=True)
        g.replay()
        if i % 500 == 0:
            torch.cuda.synchronize()
            if not torch.isfinite(grad).all() or \
                    not torch.allclose(grad, ref):
                d = (grad - ref).abs()
                print(f"MISMATCH at iter {i}: max|d|={float(d.max()):.3e} "
                      f"nonfinite={int((~torch.isfinite(grad)).sum())}",
                      flush=True)
                bad += 1
                if bad > 3:
                    sys.exit(2)
    torch.cuda.synchronize()
    print("clean:", args.iters, "replays")


if __name__ == "__main__":
    main()
