"""Learner -> actor weight publication (SURVEY.md §3c weight publish).

Double-buffered shared-memory mirror of the model parameters with a
version counter: the learner writes the inactive buffer then bumps the
version; actors copy from the active buffer and retry if the version
moved mid-copy, so they never act on torn weights and never block the
learner.
"""
from __future__ import annotations

from typing import List

import torch


class WeightMirror:
    def __init__(self, model: torch.nn.Module):
        self.version = torch.zeros(1, dtype=torch.int64)
        self.version.share_memory_()
        self._bufs: List[List[torch.Tensor]] = []
        sd = model.state_dict()
        self.keys = list(sd.keys())
        for _ in range(2):
            buf = [torch.empty_like(sd[k], device="cpu").share_memory_()
                   for k in self.keys]
            self._bufs.append(buf)
        # publish initial weights into both buffers
        for b in range(2):
            self._write(b, model)
        self.version.fill_(1)

    def _write(self, b: int, model: torch.nn.Module):
        sd = model.state_dict()
        for k, dst in zip(self.keys, self._bufs[b]):
            dst.copy_(sd[k].detach(), non_blocking=False)

    def publish(self, model: torch.nn.Module):
        """Called by the learner. Writes inactive buffer, flips version."""
        v = int(self.version.item())
        self._write((v + 1) % 2, model)
        self.version.fill_(v + 1)

    def publish_state_dict(self, sd):
        """Direct publish from a (possibly GPU) state dict — one D2H copy
        per tensor straight into the shared mirror, no CPU model hop."""
        v = int(self.version.item())
        b = (v + 1) % 2
        with torch.no_grad():
            for k, dst in zip(self.keys, self._bufs[b]):
                dst.copy_(sd[k].detach(), non_blocking=False)
        self.version.fill_(v + 1)

    def pull_into(self, model: torch.nn.Module, last_seen: int = -1) -> int:
        """Called by actors. Returns the version pulled (or last_seen if
        already current)."""
        v = int(self.version.item())
        if v == last_seen:
            return last_seen
        while True:
            src = self._bufs[v % 2]
            with torch.no_grad():
                sd = model.state_dict()
                for k, s in zip(self.keys, src):
                    sd[k].copy_(s)
            v2 = int(self.version.item())
            # version v's buffer stays intact only until the writer
            # publishes v+2 (which rewrites buffer v%2): exactly one
            # newer version is tolerable, two means our copy may be torn
            if v2 <= v + 1:
                return v
            v = v2  # writer lapped us; retry from the newest version
