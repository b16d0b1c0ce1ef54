"""bench.py is the driver contract — verify the JSON line schema and the
CPU execution path end-to-end (subprocess, tiny config)."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(420)
def test_bench_cpu_json_schema():
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--device", "cpu",
         "--steps", "4", "--warmup", "2", "--num-actors", "2",
         "--batch-size", "4", "--unroll-length", "8",
         "--envs-per-actor", "2"],
        capture_output=True, text=True, cwd=ROOT, timeout=360)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.strip().splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, f"missing contract key {key}"
    assert d["metric"].startswith("env frames/sec (learner)")
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    # value is the whole-job aggregate: steps*T*B*n_gpus / elapsed
    assert abs(d["value"] * d["ms_per_step"] / 1000.0
               - 8 * 4 * d["n_gpus"]) < 1e-6
    cfg = d["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism"):
        assert key in cfg


@pytest.mark.timeout(420)
def test_cli_train_entry():
    r = subprocess.run(
        [sys.executable, "-m", "torched_impala_amd.cli", "--env", "cartpole",
         "--device", "cpu", "--num-actors", "1", "--batch-size", "2",
         "--unroll-length", "5", "--total-steps", "40",
         "--disable-checkpoint"],
        capture_output=True, text=True, cwd=ROOT, timeout=360)
    assert r.returncode == 0, r.stderr[-2000:]


@pytest.mark.timeout(420)
def test_cli_train_deep_dmlab_cpu():
    """End-to-end train loop with the deep ResNet on the DMLab-shaped
    synthetic env (eager fp32 path on CPU; the NHWC bf16 kernel path is
    covered by test_gpu_deep.py)."""
    r = subprocess.run(
        [sys.executable, "-m", "torched_impala_amd.cli", "--env",
         "synthetic-dmlab", "--model", "deep", "--device", "cpu",
         "--num-actors", "1", "--envs-per-actor", "2", "--batch-size", "2",
         "--unroll-length", "5", "--total-steps", "20",
         "--disable-checkpoint"],
        capture_output=True, text=True, cwd=ROOT, timeout=360)
    assert r.returncode == 0, r.stderr[-2000:]


@pytest.mark.timeout(420)
def test_bench_torchrun_two_ranks_cpu():
    """The driver's multi-GPU scaling invocation, on CPU/gloo: torchrun
    2 ranks -> init_distributed -> per-rank actors -> eager learn steps
    with FlatGradReducer -> MAX-reduced timing -> rank0 JSON line."""
    import shutil
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29721", "bench.py", "--gpus", "2",
         "--device", "cpu", "--steps", "3", "--warmup", "1",
         "--num-actors", "1", "--batch-size", "2", "--unroll-length", "5",
         "--envs-per-actor", "2"],
        capture_output=True, text=True, cwd=ROOT, timeout=360)
    assert r.returncode == 0, f"stdout:\n{r.stdout[-1500:]}\nstderr:\n{r.stderr[-2500:]}"
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line (rank 0): {lines}"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 4


def test_flags_roundtrip_defaults():
    """Every CLI flag maps onto a Flags field (no silent drops)."""
    from torched_impala_amd.config import build_parser, Flags, flags_from_args
    import dataclasses
    ns = vars(build_parser().parse_args([]))
    fields = {f.name for f in dataclasses.fields(Flags)}
    unknown = set(ns) - fields
    assert not unknown, f"parser dests without Flags fields: {unknown}"
    f = flags_from_args([])
    assert f.batch_size == Flags().batch_size


def test_batch_envs_divisibility_error():
    from torched_impala_amd.config import Flags
    from torched_impala_amd.learner import create_buffers
    import numpy as np
    import pytest as _pt
    flags = Flags(batch_size=6, envs_per_actor=4)
    with _pt.raises(AssertionError):
        create_buffers(flags, (4, 84, 84), np.uint8, 6)
