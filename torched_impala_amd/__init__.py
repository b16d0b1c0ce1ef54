"""torched_impala_amd — an MI355X-native IMPALA actor-learner RL engine.

A from-scratch re-design of the capabilities of
``threewisemonkeys-as/torched_impala`` (reference repo; see SURVEY.md §0 —
the reference mount was empty, so the feature contract is BASELINE.json:5)
built MI355X-first:

* learner on PyTorch-ROCm with hand-written gfx950 HIP kernels for the
  V-trace scan, the fused policy-gradient/value/entropy loss, the LSTM cell
  and the encoder GEMMs (``torched_impala_amd/csrc``),
* CPU actor processes feeding a lock-free pinned-host trajectory ring
  buffer drained by ``hipMemcpyAsync`` on a side HIP stream,
* learner data parallelism over RCCL/xGMI with gradient all-reduce
  overlapped with backward.

Public API:
    from torched_impala_amd import vtrace, models, envs
    from torched_impala_amd.config import Flags
    from torched_impala_amd.learner import train
"""

__version__ = "0.1.0"

from torched_impala_amd.algo import vtrace  # noqa: F401
