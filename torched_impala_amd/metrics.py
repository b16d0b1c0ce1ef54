"""Structured metrics: JSONL records per learner step
interval — step, env frames/sec (the BASELINE.json:2 metric), loss
components, queue depths and per-stage timings (rollout wait, H2D copy,
forward, backward, all-reduce, optimizer)."""
from __future__ import annotations

import json
import logging
import time
from collections import deque
from typing import Optional

log = logging.getLogger(__name__)


class StageTimers:
    """Cheap wall-clock per-stage timers (CUDA-event timing is the
    profiler's job; these catch queue starvation and overlap regressions)."""

    def __init__(self):
        self.acc = {}
        self._t0 = None
        self._stage = None

    def start(self, stage: str):
        now = time.perf_counter()
        if self._stage is not None:
            self.acc[self._stage] = self.acc.get(self._stage, 0.0) + (now - self._t0)
        self._stage, self._t0 = stage, now

    def stop(self):
        self.start(None)
        self._stage = None

    def drain(self):
        out, self.acc = self.acc, {}
        out.pop(None, None)
        return out


class MetricsLogger:
    def __init__(self, jsonl_path: Optional[str] = None, window: int = 100):
        self.path = jsonl_path
        self._fh = open(jsonl_path, "a") if jsonl_path else None
        self.episode_returns = deque(maxlen=window)

    def record_episodes(self, returns):
        self.episode_returns.extend(float(r) for r in returns)

    @property
    def mean_return(self):
        if not self.episode_returns:
            return None
        return sum(self.episode_returns) / len(self.episode_returns)

    def write(self, **kv):
        if self._fh:
            self._fh.write(json.dumps(kv) + "\n")
            self._fh.flush()

    def close(self):
        if self._fh:
            self._fh.close()
