"""Find the first non-finite learner step under real training (the r12
GPU soak showed loss=nan; short tests stay finite). Reports the step at
which total_loss/pg/baseline/entropy first go non-finite, plus the last
good stats, for A/B-ing the suspected gates:

  python scripts/nan_probe.py --steps 3000                 # default path
  IMPALA_SHALLOW_MFMA=0 python scripts/nan_probe.py ...    # fp32 encoder
  python scripts/nan_probe.py --actor-jit frozen ...       # no int8 actor
  python scripts/nan_probe.py --no-graphs ...
"""
from __future__ import annotations

import argparse
import json
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=2000)
    p.add_argument("--num-actors", type=int, default=16)
    p.add_argument("--envs-per-actor", type=int, default=16)
    p.add_argument("--actor-jit", default="native-bf16")
    p.add_argument("--use-lstm", action="store_true")
    p.add_argument("--no-graphs", action="store_true")
    p.add_argument("--learning-rate", type=float, default=0.0006)
    p.add_argument("--seed", type=int, default=77)
    p.add_argument("--episode-len", type=int, default=1000)
    p.add_argument("--gradwatch", default=None,
                   help="path: on the FIRST skipped update (non-finite "
                        "grad), report per-parameter grad finiteness and "
                        "dump the batch + grads for dump_replay")
    p.add_argument("--hunt", default=None,
                   help="path prefix: device-side loss ring + rolling "
                        "staging-clone window, sparse finiteness checks "
                        "(no per-step host sync), dump the poisoned "
                        "batch on detection")
    args = p.parse_args()

    from torched_impala_amd.config import Flags
    from torched_impala_amd import learner

    flags = Flags(env="synthetic-atari", num_actors=args.num_actors,
                  envs_per_actor=args.envs_per_actor, batch_size=32,
                  unroll_length=20, total_steps=1 << 60,
                  actor_jit=args.actor_jit, use_lstm=args.use_lstm,
                  graph_capture=not args.no_graphs,
                  learning_rate=args.learning_rate,
                  disable_checkpoint=True,
                  synthetic_episode_len=args.episode_len,
                  device="cuda" if torch.cuda.is_available() else "cpu",
                  seed=args.seed)

    state = {"first_bad": None, "bad_key": None, "last_good": {}}
    ctx = None

    if args.gradwatch:
        import types
        ctx = types.SimpleNamespace()
        gw = {"last_skips": 0.0, "done": False}

        def on_step(ls, frames, stats):
            if gw["done"]:
                return
            skips = float(ctx.optimizer.skip_t.item())
            if skips > gw["last_skips"]:
                torch.cuda.synchronize()
                fg = ctx.optimizer.flat_grad
                try:
                    from torched_impala_amd.ops.shallow_mfma import STASH
                    chain = {k: [int((~torch.isfinite(v.float())).sum()),
                                 float(v.float()[torch.isfinite(v.float())]
                                       .abs().max())]
                             for k, v in STASH.items()}
                    print(json.dumps({"chain_nonfinite": chain}),
                          flush=True)
                except Exception as e:  # pragma: no cover
                    print("stash failed:", e, flush=True)
                per = []
                for n_, p_ in ctx.model.named_parameters():
                    off, num = ctx.optimizer.slices[p_]
                    sl = fg[off:off + num]
                    nf = int((~torch.isfinite(sl)).sum())
                    if nf:
                        per.append(dict(name=n_, nonfinite=nf,
                                        numel=num))
                si = (ls - 1) % 2
                blob = dict(step=ls, skips=skips,
                            per_param=per,
                            flat_param=ctx.optimizer.flat_param.cpu(),
                            flat_grad=fg.cpu(),
                            staging={k: v.cpu() for k, v in
                                     ctx.batcher.staging[si].items()},
                            stats={k: float(v) for k, v in stats.items()})
                torch.save(blob, args.gradwatch)
                print(json.dumps({"gradwatch_step": ls, "skips": skips,
                                  "bad_params": per}), flush=True)
                gw["done"] = True
                state["first_bad"] = ls
            gw["last_skips"] = skips
            if ls % 500 == 0:
                print(f"[gradwatch] step {ls} skips={skips}", flush=True)
    elif args.hunt:
        import types
        ctx = types.SimpleNamespace()
        RING = 8192
        WINDOW = 12
        hunt = {"ring": None, "clones": [], "done": False}

        def on_step(ls, frames, stats):
            # NO host sync in the hot path: device-side records only
            if hunt["ring"] is None:
                dev = stats["total_loss"].device
                hunt["ring"] = torch.zeros(RING, device=dev)
                st0 = ctx.batcher.staging[0]
                for _ in range(WINDOW):
                    hunt["clones"].append(
                        {k: torch.empty_like(v) for k, v in st0.items()})
            hunt["ring"][ls % RING].copy_(stats["total_loss"].reshape(()))
            si = (ls - 1) % 2
            cl = hunt["clones"][ls % WINDOW]
            for k, v in ctx.batcher.staging[si].items():
                cl[k].copy_(v)
            if ls % 49 == 0 and not hunt["done"]:
                lo = max(1, ls - 48)
                seg = hunt["ring"][lo:ls + 1]  # RING > max steps: no wrap
                if not bool(torch.isfinite(seg).all()):
                    torch.cuda.synchronize()
                    vals = hunt["ring"][:ls + 1].cpu()
                    bad_steps = [i for i in range(lo, ls + 1)
                                 if not math.isfinite(float(vals[i]))]
                    first = bad_steps[0]
                    blob = dict(first_bad=first, checked_at=ls,
                                ring=vals,
                                flat_param=ctx.optimizer.flat_param.cpu(),
                                flat_grad=ctx.optimizer.flat_grad.cpu(),
                                flat_sqavg=ctx.optimizer.flat_sqavg.cpu(),
                                window={})
                    for w in range(max(first - 3, lo), min(first + 2, ls + 1)):
                        blob["window"][w] = {
                            k: v.cpu() for k, v in
                            hunt["clones"][w % WINDOW].items()}
                    torch.save(blob, args.hunt)
                    print(json.dumps({"hunt_first_bad": first,
                                      "checked_at": ls,
                                      "dump": args.hunt}), flush=True)
                    hunt["done"] = True
                    state["first_bad"] = first
            if ls % 500 == 0:
                print(f"[hunt] step {ls}", flush=True)
    else:
        def on_step(ls, frames, stats):
            vals = {k: float(v) for k, v in stats.items()}
            bad = [k for k, v in vals.items() if not math.isfinite(v)]
            if bad and state["first_bad"] is None:
                state["first_bad"] = ls
                state["bad_key"] = bad
                print(json.dumps({"first_nonfinite_step": ls, "bad": bad,
                                  "stats": vals,
                                  "last_good": state["last_good"]}),
                      flush=True)
            if not bad:
                state["last_good"] = vals
            if ls % 200 == 0:
                print(f"[probe] step {ls}: " +
                      " ".join(f"{k}={v:.4g}" for k, v in vals.items()),
                      flush=True)

    out = learner.train(flags, max_learner_steps=args.steps,
                        on_step=on_step, debug_ctx=ctx)
    print(json.dumps({
        "probe": "nan", "steps": args.steps,
        "first_nonfinite_step": state["first_bad"],
        "bad_keys": state["bad_key"],
        "final": {k: v for k, v in out.items()
                  if isinstance(v, (int, float))},
        "config": {"actor_jit": args.actor_jit, "lstm": args.use_lstm,
                   "graphs": not args.no_graphs,
                   "mfma": os.environ.get("IMPALA_SHALLOW_MFMA", "default"),
                   "lr": args.learning_rate},
    }), flush=True)


if __name__ == "__main__":
    main()
