"""Flag surface of the engine (the CLI API contract).

Mirrors the canonical argparse surface of small PyTorch IMPALA
implementations (env id, num_actors, total_steps,
batch_size, unroll_length, lr, entropy/baseline costs, discount, grad
clip, use_lstm, device, savedir, mode), plus the MI355X-specific knobs
(num_learner_gpus is implied by torchrun world size; kernel on/off escape
hatches for A/B parity testing; synthetic-env controls).
"""
from __future__ import annotations

import argparse
import dataclasses
from typing import Optional


@dataclasses.dataclass
class Flags:
    # --- experiment ---
    env: str = "synthetic-atari"       # synthetic-atari | synthetic-dmlab | cartpole
    mode: str = "train"                # train | test
    xpid: Optional[str] = None         # experiment id (run dir name)
    savedir: str = "~/impala_runs"
    total_steps: int = 100000          # env (agent) frames to train for
    disable_checkpoint: bool = False
    checkpoint_interval_s: float = 600.0

    # --- actor/learner topology ---
    num_actors: int = 4                # CPU actor processes per learner GPU
    batch_size: int = 8                # unrolls per learner batch (B)
    unroll_length: int = 20            # T
    num_buffers: Optional[int] = None  # ring slots; default max(2*B, B+num_actors)
    envs_per_actor: int = 1            # batched envs inside one actor process
    actor_threads: int = 0             # CPU threads per actor; 0 = auto
                                       # (cores/num_actors, capped at 8)
    actor_jit: str = "native"          # off | safe | frozen | native |
                                       # native-bf16 (AVX512-BF16 conv/fc
                                       # stages, ~2x MAC rate; behavior
                                       # logits are the bf16 policy's --
                                       # V-trace corrects staleness and
                                       # precision alike)
    actor_jit_refresh: int = 4

    # --- model ---
    use_lstm: bool = False
    model: str = "auto"                # auto | mlp | shallow | deep

    # --- loss / algorithm (IMPALA paper App. G canonical values) ---
    entropy_cost: float = 0.01
    baseline_cost: float = 0.5
    discounting: float = 0.99
    reward_clipping: str = "abs_one"   # abs_one | none
    # benchmarking: block the learner until the rollout ring is fully
    # populated (every actor warmed through >=1 claim) so short timed
    # windows measure steady state, not the actor spawn ramp
    ramp_ready: bool = False
    clip_rho_threshold: float = 1.0
    clip_c_threshold: float = 1.0
    # rho-bar for the PG advantage; None => same as clip_rho_threshold
    # (the HIP scan and eager paths are A/B-parity-tested on this value)
    clip_pg_rho_threshold: float = None
    lambda_: float = 1.0

    # --- optimizer (RMSProp, paper defaults) ---
    learning_rate: float = 0.0006
    alpha: float = 0.99               # RMSProp smoothing
    momentum: float = 0.0
    epsilon: float = 0.01
    grad_norm_clipping: float = 40.0

    # --- device / kernels ---
    device: str = "auto"              # auto | cuda | cpu  ("cuda" is ROCm/HIP)
    use_hip_kernels: bool = True      # escape hatch: False = eager PyTorch ops
    graph_capture: bool = True        # hipGraph-capture the learn step
    compute_dtype: str = "fp32"       # fp32 | bf16 (encoder GEMM compute)

    # --- synthetic env shapes ---
    synthetic_frame_hw: Optional[str] = None  # e.g. "84x84x4" / "72x96x3"
    synthetic_num_actions: int = 6
    synthetic_episode_len: int = 1000

    # --- misc ---
    seed: int = 1234
    metrics_jsonl: Optional[str] = None
    weight_sync_interval: int = 1     # learner steps between weight publishes

    @property
    def run_device(self) -> str:
        if self.device != "auto":
            return self.device
        import torch
        return "cuda" if torch.cuda.is_available() else "cpu"


def _add_bool(parser: argparse.ArgumentParser, name: str, default: bool, help_: str = ""):
    parser.add_argument(f"--{name}", dest=name.replace("-", "_"),
                        action="store_true", default=default, help=help_)
    parser.add_argument(f"--no-{name}", dest=name.replace("-", "_"),
                        action="store_false")


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="torched_impala_amd IMPALA engine")
    f = Flags()
    p.add_argument("--env", default=f.env)
    p.add_argument("--mode", default=f.mode, choices=["train", "test"])
    p.add_argument("--xpid", default=None)
    p.add_argument("--savedir", default=f.savedir)
    p.add_argument("--total-steps", type=int, default=f.total_steps)
    _add_bool(p, "disable-checkpoint", f.disable_checkpoint)
    p.add_argument("--num-actors", type=int, default=f.num_actors)
    p.add_argument("--batch-size", type=int, default=f.batch_size)
    p.add_argument("--unroll-length", type=int, default=f.unroll_length)
    p.add_argument("--num-buffers", type=int, default=None)
    p.add_argument("--envs-per-actor", type=int, default=f.envs_per_actor)
    p.add_argument("--actor-threads", type=int, default=f.actor_threads)
    p.add_argument("--actor-jit", default=f.actor_jit,
                   choices=["off", "safe", "frozen", "native",
                            "native-bf16"])
    p.add_argument("--actor-jit-refresh", type=int, default=f.actor_jit_refresh)
    _add_bool(p, "use-lstm", f.use_lstm)
    p.add_argument("--model", default=f.model,
                   choices=["auto", "mlp", "shallow", "deep"])
    p.add_argument("--entropy-cost", type=float, default=f.entropy_cost)
    p.add_argument("--baseline-cost", type=float, default=f.baseline_cost)
    p.add_argument("--discounting", type=float, default=f.discounting)
    p.add_argument("--reward-clipping", default=f.reward_clipping,
                   choices=["abs_one", "none"])
    p.add_argument("--clip-rho-threshold", type=float,
                   default=f.clip_rho_threshold)
    p.add_argument("--clip-c-threshold", type=float,
                   default=f.clip_c_threshold)
    p.add_argument("--clip-pg-rho-threshold", type=float,
                   default=f.clip_pg_rho_threshold)
    p.add_argument("--lambda", dest="lambda_", type=float, default=f.lambda_)
    p.add_argument("--learning-rate", type=float, default=f.learning_rate)
    p.add_argument("--alpha", type=float, default=f.alpha)
    p.add_argument("--momentum", type=float, default=f.momentum)
    p.add_argument("--epsilon", type=float, default=f.epsilon)
    p.add_argument("--grad-norm-clipping", type=float, default=f.grad_norm_clipping)
    p.add_argument("--device", default=f.device)
    _add_bool(p, "use-hip-kernels", f.use_hip_kernels)
    _add_bool(p, "graph-capture", f.graph_capture)
    p.add_argument("--compute-dtype", default=f.compute_dtype, choices=["fp32", "bf16"])
    p.add_argument("--synthetic-frame-hw", default=None)
    p.add_argument("--synthetic-num-actions", type=int, default=f.synthetic_num_actions)
    p.add_argument("--synthetic-episode-len", type=int, default=f.synthetic_episode_len)
    p.add_argument("--seed", type=int, default=f.seed)
    p.add_argument("--metrics-jsonl", default=None)
    p.add_argument("--checkpoint-interval-s", type=float,
                   default=f.checkpoint_interval_s)
    p.add_argument("--weight-sync-interval", type=int, default=f.weight_sync_interval)
    return p


def flags_from_args(argv=None) -> Flags:
    ns = build_parser().parse_args(argv)
    d = vars(ns)
    # argparse dest for lambda_ etc. are already snake_case
    known = {fld.name for fld in dataclasses.fields(Flags)}
    return Flags(**{k: v for k, v in d.items() if k in known})
