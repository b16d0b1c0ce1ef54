"""Learner -> actor weight publication (SURVEY.md §3c weight publish).

Double-buffered shared-memory mirror of the model parameters with a
version counter plus a per-buffer seqlock: the learner marks the
inactive buffer's sequence odd, writes it, marks it even, then bumps
the version; actors copy from the active buffer only at an even,
unchanged-across-the-copy sequence. Readers never act on torn weights
and never block the learner.
"""
from __future__ import annotations

import time
from typing import List

import torch


class WeightMirror:
    """Seqlock protocol: each buffer has a write-sequence counter that the
    writer makes ODD before touching the buffer and EVEN after.  A reader
    copies only when the sequence is even and unchanged across its copy —
    a version-counter check alone cannot protect against the writer
    re-using the same buffer two publishes later (observed as a
    two-version tear under a fast writer)."""

    def __init__(self, model: torch.nn.Module):
        self.version = torch.zeros(1, dtype=torch.int64)
        self.version.share_memory_()
        self.seq = torch.zeros(2, dtype=torch.int64)   # per-buffer seqlock
        self.seq.share_memory_()
        self._bufs: List[List[torch.Tensor]] = []
        sd = model.state_dict()
        self.keys = list(sd.keys())
        for _ in range(2):
            buf = [torch.empty_like(sd[k], device="cpu").share_memory_()
                   for k in self.keys]
            self._bufs.append(buf)
        # publish initial weights into both buffers (no readers yet)
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
        b = (v + 1) % 2
        self.seq[b] += 1               # odd: writer active in buffer b
        self._write(b, model)
        self.seq[b] += 1               # even: buffer b consistent
        self.version.fill_(v + 1)

    def publish_state_dict(self, sd):
        """Direct publish from a (possibly GPU) state dict — one D2H copy
        per tensor straight into the shared mirror, no CPU model hop."""
        v = int(self.version.item())
        b = (v + 1) % 2
        self.seq[b] += 1
        with torch.no_grad():
            for k, dst in zip(self.keys, self._bufs[b]):
                dst.copy_(sd[k].detach(), non_blocking=False)
        self.seq[b] += 1
        self.version.fill_(v + 1)

    def pull_into(self, model: torch.nn.Module, last_seen: int = -1) -> int:
        """Called by actors. Returns the version pulled (or last_seen if
        already current)."""
        if int(self.version.item()) == last_seen:
            return last_seen
        while True:
            v = int(self.version.item())
            b = v % 2
            s1 = int(self.seq[b].item())
            if s1 & 1:                 # writer mid-overwrite of this buffer
                time.sleep(2e-4)
                continue
            src = self._bufs[b]
            with torch.no_grad():
                sd = model.state_dict()
                for k, s in zip(self.keys, src):
                    sd[k].copy_(s)
            if int(self.seq[b].item()) == s1:
                return v
            # the writer re-entered buffer b during our copy; retry
