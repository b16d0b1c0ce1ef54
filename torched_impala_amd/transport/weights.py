"""Learner -> actor weight publication (SURVEY.md §3c weight publish).

Double-buffered shared-memory mirror of the model parameters with a
version counter plus a per-buffer seqlock: the learner marks the
inactive buffer's sequence odd, writes it, marks it even, then bumps
the version; actors copy from the active buffer only at an even,
unchanged-across-the-copy sequence. Readers never act on torn weights
and never block the learner.

Two publish paths:

* ``publish_state_dict`` — synchronous per-tensor copy (CPU learner,
  resume, tests).
* ``publish_flat_async`` — the GPU hot path: when
  the mirror was built with the FlatRMSprop layout, each buffer is ONE
  flat pinned shm tensor with per-key views, and a publish is a single
  device->device snapshot (compute-stream ordered, so the next step
  can't tear it) followed by ONE async D2H on a dedicated stream; the
  seqlock flip happens when the copy event completes (polled from the
  learner loop). Host-blocking cost per step: an event query.
"""
from __future__ import annotations

import time
from typing import Dict, List, Optional, Tuple

import torch


class WeightMirror:
    """Seqlock protocol: each buffer has a write-sequence counter that the
    writer makes ODD before touching the buffer and EVEN after.  A reader
    copies only when the sequence is even and unchanged across its copy —
    a version-counter check alone cannot protect against the writer
    re-using the same buffer two publishes later (observed as a
    two-version tear under a fast writer)."""

    def __init__(self, model: torch.nn.Module,
                 flat_layout: Optional[Dict[str, Tuple[int, int]]] = None):
        """flat_layout: {state_dict key: (offset, numel)} into a single
        flat fp32 arena (the FlatRMSprop layout). Keys not in the layout
        (non-param buffers) get standalone tensors as before."""
        self.version = torch.zeros(1, dtype=torch.int64)
        self.version.share_memory_()
        self.seq = torch.zeros(2, dtype=torch.int64)   # per-buffer seqlock
        self.seq.share_memory_()
        sd = model.state_dict()
        self.keys = list(sd.keys())
        self.layout = None
        self.flats: List[torch.Tensor] = []
        self._bufs: List[List[torch.Tensor]] = []
        if flat_layout is not None and all(
                k in flat_layout and sd[k].dtype == torch.float32
                for k in self.keys):
            self.layout = dict(flat_layout)
            total = max(off + n for off, n in flat_layout.values())
            for _ in range(2):
                flat = torch.empty(total, dtype=torch.float32)
                flat.share_memory_()
                self.flats.append(flat)
                buf = []
                for k in self.keys:
                    off, n = flat_layout[k]
                    assert n == sd[k].numel(), f"layout mismatch for {k}"
                    buf.append(flat[off:off + n].view(sd[k].shape))
                self._bufs.append(buf)
        else:
            for _ in range(2):
                buf = [torch.empty_like(sd[k], device="cpu").share_memory_()
                       for k in self.keys]
                self._bufs.append(buf)
        # async-publish state (learner process only; never pickled)
        self._pending = None          # (buffer index, version at begin)
        self._pending_ev = None
        self._pub_stream = None
        self._scratch = None
        # publish initial weights into both buffers (no readers yet)
        for b in range(2):
            self._write(b, model)
        self.version.fill_(1)

    def __getstate__(self):
        d = self.__dict__.copy()
        # device-side publish machinery must not cross the process fork
        d["_pending"] = None
        d["_pending_ev"] = None
        d["_pub_stream"] = None
        d["_scratch"] = None
        return d

    def pin(self):
        """hipHostRegister the flat buffers for async D2H (learner only)."""
        from torched_impala_amd import ops
        for f in self.flats:
            ops.pin_host_memory(f)

    def _write(self, b: int, model: torch.nn.Module):
        sd = model.state_dict()
        for k, dst in zip(self.keys, self._bufs[b]):
            dst.copy_(sd[k].detach(), non_blocking=False)

    # ------------------------------------------------------------------
    # synchronous publishes
    # ------------------------------------------------------------------
    def publish(self, model: torch.nn.Module):
        """Called by the learner. Writes inactive buffer, flips version."""
        self._finish_pending(wait=True)
        v = int(self.version.item())
        b = (v + 1) % 2
        self.seq[b] += 1               # odd: writer active in buffer b
        self._write(b, model)
        self.seq[b] += 1               # even: buffer b consistent
        self.version.fill_(v + 1)

    def publish_state_dict(self, sd):
        """Direct publish from a (possibly GPU) state dict — one D2H copy
        per tensor straight into the shared mirror, no CPU model hop."""
        self._finish_pending(wait=True)
        v = int(self.version.item())
        b = (v + 1) % 2
        self.seq[b] += 1
        with torch.no_grad():
            for k, dst in zip(self.keys, self._bufs[b]):
                dst.copy_(sd[k].detach(), non_blocking=False)
        self.seq[b] += 1
        self.version.fill_(v + 1)

    # ------------------------------------------------------------------
    # async flat publish (GPU hot path)
    # ------------------------------------------------------------------
    def publish_flat_async(self, flat_param: torch.Tensor) -> bool:
        """One compute-stream D2D snapshot + one async D2H; seqlock flip
        deferred to ``poll_publish``. Returns False (skipped) if the
        previous async publish is still in flight — the next call
        publishes fresher weights anyway. Requires the flat layout."""
        assert self.layout is not None, "mirror not built with flat layout"
        if not flat_param.is_cuda:
            # CPU learner: one flat copy, immediate flip
            self._finish_pending(wait=True)
            v = int(self.version.item())
            b = (v + 1) % 2
            self.seq[b] += 1
            self.flats[b].copy_(flat_param)
            self.seq[b] += 1
            self.version.fill_(v + 1)
            return True
        if self._pending is not None:
            if not self._pending_ev.query():
                return False
            self._complete()
        if self._pub_stream is None:
            self._pub_stream = torch.cuda.Stream()
            self._scratch = torch.empty_like(flat_param)
        v = int(self.version.item())
        b = (v + 1) % 2
        self.seq[b] += 1               # odd: writer active in buffer b
        # snapshot on the COMPUTE stream: ordered after this step's
        # optimizer update and before the next step's — no torn mix of
        # step N and N+1 even though the D2H below overlaps compute.
        self._scratch.copy_(flat_param, non_blocking=True)
        self._pub_stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(self._pub_stream):
            self.flats[b].copy_(self._scratch, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record(self._pub_stream)
        self._pending = (b, v)
        self._pending_ev = ev
        return True

    def poll_publish(self):
        """Flip the seqlock/version of a completed async publish (cheap
        event query; call once per learner step)."""
        if self._pending is not None and self._pending_ev.query():
            self._complete()

    def _complete(self):
        b, v = self._pending
        self.seq[b] += 1               # even: buffer b consistent
        self.version.fill_(v + 1)
        self._pending = None
        self._pending_ev = None

    def _finish_pending(self, wait: bool):
        if self._pending is None:
            return
        if wait:
            self._pending_ev.synchronize()
        self._complete()

    # ------------------------------------------------------------------
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
