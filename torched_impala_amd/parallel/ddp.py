"""Learner data parallelism: bucketed gradient all-reduce over RCCL/xGMI,
overlapped with backward (BASELINE.json:5).

Design notes (MI355X): xGMI is 7 point-to-point links x ~153 GB/s per
GPU; ring all-reduce is per-link bound, but the IMPALA models are tiny
(shallow net ~2-6 MB fp32 of grads) so latency dominates — buckets are
therefore FEW and LARGE (default one ~25 MB ceiling bucket per model
unless the model exceeds it), launched per-bucket from
post-accumulate-grad hooks as backward produces them, on RCCL's own
stream so the reduction overlaps the remaining backward compute.

Backend "nccl" IS RCCL on ROCm; tests run the same code on gloo/CPU
(world_size 2) — SURVEY.md §4.5.
"""
from __future__ import annotations

import logging
import os
from typing import List, Optional

import torch
import torch.distributed as dist

log = logging.getLogger(__name__)


def init_distributed(device: str) -> tuple[int, int]:
    """torchrun-style init from env vars. Returns (rank, world_size)."""
    if "RANK" not in os.environ or int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        return 0, 1
    backend = "nccl" if device == "cuda" else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29533")
    if not dist.is_initialized():
        dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    if device == "cuda":
        local = int(os.environ.get("LOCAL_RANK", rank))
        # modulo lets DP logic be exercised with ranks > GPUs (e.g. two
        # RCCL ranks sharing the single gpurun GPU — SURVEY.md §4.5)
        torch.cuda.set_device(local % max(1, torch.cuda.device_count()))
    return rank, dist.get_world_size()


class FlatGradReducer:
    """Zero-copy bucketed all-reduce over a FlatRMSprop gradient arena.

    Buckets are contiguous slices of ``optimizer.flat_grad`` aligned to
    parameter boundaries in flat (reverse-parameter ≈ backward) order;
    post-accumulate-grad hooks launch each bucket's async all-reduce the
    moment its last grad lands, overlapping the rest of backward. No
    copy-in/copy-out; the 1/world mean is folded into the optimizer's
    device-side scale (FlatRMSprop(world_size=...)).
    """

    def __init__(self, model, optimizer, bucket_bytes: int = 25 << 20,
                 process_group=None):
        del model
        self.group = process_group
        self.opt = optimizer
        self.world_size = dist.get_world_size(process_group) if (
            dist.is_available() and dist.is_initialized()) else 1
        self.enabled = self.world_size > 1
        self.buckets = []
        self._param_bucket = {}
        if not self.enabled:
            return
        flat = optimizer.flat_grad
        cur_params, cur_start, cur_end = [], 0, 0
        for p in optimizer._order:
            off, n = optimizer.slices[p]
            cur_params.append(p)
            cur_end = off + n
            if (cur_end - cur_start) * 4 >= bucket_bytes:
                self.buckets.append(
                    dict(params=list(cur_params), view=flat[cur_start:cur_end],
                         ready=0, work=None))
                cur_params, cur_start = [], cur_end
        if cur_params:
            self.buckets.append(
                dict(params=list(cur_params), view=flat[cur_start:cur_end],
                     ready=0, work=None))
        for b in self.buckets:
            for p in b["params"]:
                self._param_bucket[p] = b
                p.register_post_accumulate_grad_hook(self._hook)
        log.info("FlatGradReducer: %d buckets, world=%d", len(self.buckets),
                 self.world_size)

    def _hook(self, p):
        b = self._param_bucket[p]
        b["ready"] += 1
        if b["ready"] == len(b["params"]):
            b["work"] = dist.all_reduce(b["view"], op=dist.ReduceOp.SUM,
                                        group=self.group, async_op=True)

    def prepare(self):
        if not self.enabled:
            return
        for b in self.buckets:
            b["ready"], b["work"] = 0, None

    def finalize(self):
        if not self.enabled:
            return
        for b in self.buckets:
            if b["work"] is None:  # keep ranks collectively consistent
                b["work"] = dist.all_reduce(b["view"], op=dist.ReduceOp.SUM,
                                            group=self.group, async_op=True)
        for b in self.buckets:
            b["work"].wait()


class _Bucket:
    __slots__ = ("params", "numel", "flat", "work", "ready")

    def __init__(self, params: List[torch.nn.Parameter], device, dtype):
        self.params = params
        self.numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(self.numel, device=device, dtype=dtype)
        self.work = None
        self.ready = 0


class GradReducer:
    """Bucketed, backward-overlapped gradient all-reduce.

    Usage per step:
        reducer.prepare()          # before backward
        loss.backward()            # hooks fire per-bucket all_reduce
        reducer.finalize()         # wait works, write averaged grads back
    No-op when world_size == 1.
    """

    def __init__(self, model: torch.nn.Module, bucket_bytes: int = 25 << 20,
                 process_group=None):
        self.model = model
        self.group = process_group
        self.world_size = dist.get_world_size(process_group) if (
            dist.is_available() and dist.is_initialized()) else 1
        self.enabled = self.world_size > 1
        self.buckets: List[_Bucket] = []
        self._param_bucket = {}
        if not self.enabled:
            return
        params = [p for p in model.parameters() if p.requires_grad]
        # reverse order approximates backward completion order
        params = list(reversed(params))
        device = params[0].device
        dtype = params[0].dtype
        cur, cur_bytes = [], 0
        for p in params:
            cur.append(p)
            cur_bytes += p.numel() * p.element_size()
            if cur_bytes >= bucket_bytes:
                self.buckets.append(_Bucket(cur, device, dtype))
                cur, cur_bytes = [], 0
        if cur:
            self.buckets.append(_Bucket(cur, device, dtype))
        for b in self.buckets:
            for p in b.params:
                self._param_bucket[p] = b
                p.register_post_accumulate_grad_hook(self._hook)
        log.info("GradReducer: %d buckets over %d params, world=%d",
                 len(self.buckets), len(params), self.world_size)

    def _hook(self, p: torch.nn.Parameter):
        b = self._param_bucket[p]
        b.ready += 1
        if b.ready == len(b.params):
            self._launch(b)

    def _launch(self, b: _Bucket):
        off = 0
        for p in b.params:
            n = p.numel()
            b.flat[off:off + n].copy_(p.grad.detach().view(-1))
            off += n
        b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                 group=self.group, async_op=True)

    def prepare(self):
        if not self.enabled:
            return
        for b in self.buckets:
            b.ready = 0
            b.work = None

    def finalize(self):
        if not self.enabled:
            return
        inv = 1.0 / self.world_size
        for b in self.buckets:
            if b.work is None:
                # params whose grads never materialized this step: reduce zeros
                # so ranks stay collectively consistent
                b.flat.zero_()
                off = 0
                for p in b.params:
                    n = p.numel()
                    if p.grad is not None:
                        b.flat[off:off + n].copy_(p.grad.detach().view(-1))
                    off += n
                b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                         group=self.group, async_op=True)
        for b in self.buckets:
            b.work.wait()
            off = 0
            for p in b.params:
                n = p.numel()
                if p.grad is None:
                    p.grad = torch.zeros_like(p)
                p.grad.detach().view(-1).copy_(b.flat[off:off + n]).mul_(inv)
                off += n
