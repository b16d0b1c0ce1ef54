"""Graph-vs-eager lockstep drift on a FIXED batch (no actors): the r13
matrix showed MFMA+graphs -> NaN at ~1-2k steps while MFMA+eager and
fp32+graphs stay clean. Replays and eager steps from identical initial
weights on the same staged batch must track each other (atomics-level
noise only); wild divergence or non-finite values at step K localises
the replay bug. Reports per-parameter grad/param health at divergence.
"""
from __future__ import annotations

import argparse
import json
import math
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


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=400)
    p.add_argument("--use-lstm", action="store_true")
    args = p.parse_args()
    device = torch.device("cuda", 0)
    obs, A = (4, 84, 84), 6
    flags = Flags(env="synthetic-atari", batch_size=32, unroll_length=20,
                  use_lstm=args.use_lstm, total_steps=1 << 50)

    torch.manual_seed(42)
    model_g = build_model(flags, obs, A).to(device)
    opt_g, sched_g = L.build_optimizer(flags, model_g)
    torch.manual_seed(42)
    model_e = build_model(flags, obs, A).to(device)
    opt_e, sched_e = L.build_optimizer(flags, model_e)
    assert float((opt_g.flat_param - opt_e.flat_param).abs().max()) == 0.0

    st = staging(flags, obs, A, device, seed=7)
    stub = Stub(flags, st)
    batch = stub._to_time_major(st)
    graphed = L.GraphedStep(flags, model_g, opt_g, stub)

    name_of = []
    for n_, p_ in model_g.named_parameters():
        off, num = opt_g.slices[p_]
        name_of.append((n_, off, num))

    for step in range(args.steps):
        graphed.step(0, sched_g)
        L.learn_step(flags, model_e, dict(batch), opt_e, sched_e, None)
        if step % 10 == 0 or step < 5:
            torch.cuda.synchronize()
            pg, pe = opt_g.flat_param, opt_e.flat_param
            d = (pg - pe).abs()
            rel = float(d.max() / (pe.abs().max() + 1e-12))
            gfin = bool(torch.isfinite(pg).all())
            efin = bool(torch.isfinite(pe).all())
            print(f"[drift] step {step}: max|d|={float(d.max()):.3e} "
                  f"rel={rel:.3e} g_finite={gfin} e_finite={efin}",
                  flush=True)
            if not gfin or not efin or rel > 0.5:
                per = []
                for n_, off, num in name_of:
                    dg = pg[off:off + num]
                    de = pe[off:off + num]
                    per.append(dict(
                        name=n_,
                        g_fin=bool(torch.isfinite(dg).all()),
                        e_fin=bool(torch.isfinite(de).all()),
                        maxd=float((dg - de).abs().max()),
                        g_absmax=float(dg.abs().max()),
                        e_absmax=float(de.abs().max())))
                print(json.dumps({"diverged_at": step, "per_param": per}),
                      flush=True)
                sys.exit(2)
    print(json.dumps({"drift_probe": "clean", "steps": args.steps}))


def stress(args):
    """--stress mode: graphed-only replay with FRESH RANDOM batch data
    written into the fixed staging buffers before every replay (the real
    Batcher's pattern), lr=0 so weights must stay bitwise-frozen; any
    non-finite loss/param is memory corruption in the replayed step.
    --buffers 2 reproduces train()'s double-buffer alternation (capture
    g0 -> replay -> capture g1 under IMPALA_GRAPH_LAZY_CAPTURE=1 — the
    shared-pool contract violation)."""
    device = torch.device("cuda", 0)
    obs, A = (4, 84, 84), 6
    flags = Flags(env="synthetic-atari", batch_size=32, unroll_length=20,
                  use_lstm=args.use_lstm, total_steps=1 << 50,
                  learning_rate=0.0)
    torch.manual_seed(42)
    model = build_model(flags, obs, A).to(device)
    opt, sched = L.build_optimizer(flags, model)
    p0 = opt.flat_param.clone()
    st0 = staging(flags, obs, A, device, seed=7)
    stub = Stub(flags, st0)
    if args.buffers == 2:
        stub.staging[1] = staging(flags, obs, A, device, seed=8)
    graphed = L.GraphedStep(flags, model, opt, stub)
    g = torch.Generator(device="cuda").manual_seed(123)
    T = flags.unroll_length
    for step in range(args.steps):
        st = stub.staging[step % args.buffers]
        # refresh the staging buffers in place (what Batcher's H2D does)
        st["frame"].random_(0, 256, generator=g)
        st["reward"].normal_(generator=g)
        st["done"].bernoulli_(0.05, generator=g)
        st["action"].random_(0, A, generator=g)
        st["last_action"].random_(0, A, generator=g)
        st["behavior_logits"].normal_(generator=g)
        stats = graphed.step(step % args.buffers, sched)
        if step % 100 == 0 or step == args.steps - 1:
            torch.cuda.synchronize()
            loss = float(stats["total_loss"])
            pfin = bool(torch.isfinite(opt.flat_param).all())
            gfin = bool(torch.isfinite(opt.flat_grad).all())
            pmax = float((opt.flat_param - p0).abs().max())
            print(f"[stress] step {step}: loss={loss:.4g} param_fin={pfin} "
                  f"grad_fin={gfin} param_drift={pmax:.3e}", flush=True)
            if not math.isfinite(loss) or not pfin or pmax > 0:
                # drill: which grad slices are bad
                per = []
                for n_, p_ in model.named_parameters():
                    off, num = opt.slices[p_]
                    gsl = opt.flat_grad[off:off + num]
                    psl = opt.flat_param[off:off + num]
                    per.append(dict(
                        name=n_, g_fin=bool(torch.isfinite(gsl).all()),
                        p_fin=bool(torch.isfinite(psl).all()),
                        g_absmax=float(gsl.abs().max()),
                        p_drift=float((psl - p0[off:off + num]).abs().max())))
                print(json.dumps({"corrupt_at": step, "per_param": per}),
                      flush=True)
                sys.exit(2)
    print(json.dumps({"stress": "clean", "steps": args.steps}))


_orig_main = main


def main():  # noqa: F811
    import argparse as _ap
    pre = _ap.ArgumentParser(add_help=False)
    pre.add_argument("--stress", action="store_true")
    pre.add_argument("--steps", type=int, default=400)
    pre.add_argument("--use-lstm", action="store_true")
    pre.add_argument("--buffers", type=int, default=1, choices=[1, 2])
    known, _ = pre.parse_known_args()
    if known.stress:
        stress(known)
    else:
        _orig_main()


if __name__ == "__main__":
    main()
