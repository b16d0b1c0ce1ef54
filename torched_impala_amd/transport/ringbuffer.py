"""Shared-memory trajectory ring buffer (actor -> learner transport).

Data plane: one shared arena tensor per rollout key, shape
``(num_slots, T+1, ...)`` — actors write whole unroll slots, the learner
drains them slot-contiguously with ``hipMemcpyAsync`` on a side stream
after registering the arenas page-locked (``ops.pin_host_memory``), so
rollout copy overlaps the learner forward/backward (BASELINE.json:5).

Control plane: two lock-free MPMC index queues (free list / full list)
implemented in C++ over shared memory (csrc/ring.cpp), with a
multiprocessing.Queue fallback when the extension is not built.
Correctness invariants (no lost/duplicated slots, tear-freedom via
per-cell sequence numbers) are stress-tested in tests/test_ringbuffer.py.
"""
from __future__ import annotations

import logging
from typing import Dict, Tuple

import torch

log = logging.getLogger(__name__)


def _load_ring_ext():
    try:
        from torched_impala_amd.ops import ext_cpu  # built extension loader
        return ext_cpu()
    except Exception as e:  # pragma: no cover
        log.warning("ring C++ extension unavailable (%s); using mp.Queue fallback", e)
        return None


class SlotQueue:
    """MPMC queue of slot indices, shared across processes.

    Lock-free C++ implementation when the extension is built; otherwise a
    multiprocessing.Queue with the same API. Both are picklable into
    spawn-context children.
    """

    def __init__(self, capacity: int, ctx=None, force_fallback: bool = False):
        cap = 1
        while cap < max(2, capacity):
            cap *= 2
        self.capacity = cap
        ext = None if force_fallback else _load_ring_ext()
        if ext is not None:
            n = int(ext.words_needed(cap))
            self._buf = torch.zeros(n, dtype=torch.int64)
            self._buf.share_memory_()
            ext.init(self._buf, cap)
            self._mpq = None
        else:
            import torch.multiprocessing as mp
            ctx = ctx or mp.get_context("spawn")
            self._buf = None
            self._mpq = ctx.Queue(maxsize=cap)

    # NOTE: the ext module handle is NOT pickled; children re-load it.
    def __getstate__(self):
        return {"capacity": self.capacity, "_buf": self._buf, "_mpq": self._mpq}

    def __setstate__(self, state):
        self.__dict__.update(state)

    @property
    def _ext(self):
        ext = _load_ring_ext()
        if ext is None:
            raise RuntimeError("ring extension missing in child process")
        return ext

    def push(self, value: int) -> bool:
        if self._buf is not None:
            return bool(self._ext.push(self._buf, int(value)))
        try:
            self._mpq.put_nowait(int(value))
            return True
        except Exception:
            return False

    def pop(self) -> int:
        if self._buf is not None:
            return int(self._ext.pop(self._buf))
        try:
            return int(self._mpq.get_nowait())
        except Exception:
            return -1

    def pop_wait(self, timeout_ms: float = -1.0) -> int:
        """-1 on timeout, -2 on abort."""
        if self._buf is not None:
            return int(self._ext.pop_wait(self._buf, float(timeout_ms)))
        import queue as _q
        try:
            timeout = None if timeout_ms < 0 else timeout_ms / 1000.0
            return int(self._mpq.get(timeout=timeout))
        except _q.Empty:
            return -1

    def abort(self):
        if self._buf is not None:
            self._ext.abort(self._buf)
        else:
            # fallback: push sentinels so waiters wake
            for _ in range(self.capacity):
                try:
                    self._mpq.put_nowait(-2)
                except Exception:
                    break

    def size_approx(self) -> int:
        if self._buf is not None:
            return int(self._ext.size_approx(self._buf))
        try:
            return self._mpq.qsize()
        except Exception:
            return 0


class TrajectoryBuffers:
    """Shared rollout arenas: key -> (num_slots, T+1, ...) tensor.

    Row semantics (monobeast-compatible):
      row t = (frame_t, reward_t, done_t) as produced by the env at step t,
      plus (action_t, behavior_logits_t) chosen by the actor AT frame_t.
    Row 0 of unroll m+1 equals row T of unroll m (one overlap row), so the
    learner trains each transition exactly once and bootstraps at row T.
    """

    def __init__(self, unroll_length: int, obs_shape: Tuple[int, ...],
                 obs_dtype: torch.dtype, num_actions: int, num_slots: int,
                 lstm_hidden: int = 0, envs_per_slot: int = 1):
        T = unroll_length
        n = envs_per_slot
        self.unroll_length = T
        self.num_slots = num_slots
        self.num_actions = num_actions
        self.lstm_hidden = lstm_hidden
        self.envs_per_slot = n
        specs = dict(
            frame=((T + 1, n, *obs_shape), obs_dtype),
            reward=((T + 1, n), torch.float32),
            done=((T + 1, n), torch.bool),
            action=((T + 1, n), torch.int64),
            behavior_logits=((T + 1, n, num_actions), torch.float32),
            last_action=((T + 1, n), torch.int64),
            episode_return=((T + 1, n), torch.float32),
            episode_step=((T + 1, n), torch.int32),
        )
        self.arenas: Dict[str, torch.Tensor] = {}
        for key, (shape, dtype) in specs.items():
            t = torch.zeros((num_slots, *shape), dtype=dtype)
            t.share_memory_()
            self.arenas[key] = t
        if lstm_hidden > 0:
            t = torch.zeros((num_slots, 2, n, lstm_hidden), dtype=torch.float32)
            t.share_memory_()
            self.arenas["initial_core_state"] = t
        self._pinned = False

    def slot(self, m: int) -> Dict[str, torch.Tensor]:
        return {k: v[m] for k, v in self.arenas.items()}

    def nbytes(self) -> int:
        return sum(v.numel() * v.element_size() for v in self.arenas.values())

    def pin(self):
        """hipHostRegister every arena in THIS process (the learner), so
        hipMemcpyAsync drains are true async DMA. No-op without a GPU."""
        if self._pinned or not torch.cuda.is_available():
            return
        from torched_impala_amd import ops
        for v in self.arenas.values():
            ops.pin_host_memory(v)
        self._pinned = True
