"""Flat-arena RMSProp with fused global-norm clipping.

All parameters, grads and RMSProp state live in single contiguous fp32
buffers; ``p.data`` / ``p.grad`` are views into them (ordered by REVERSE
parameter order ≈ backward completion order, so DP buckets fill
contiguously). Consequences:

* optimizer + clip = 2 HIP kernels total (csrc/optim_kernels.hip), with
  lr / DP-scale as device scalars -> hipGraph-capturable;
* data-parallel gradient reduction = zero-copy all-reduce of (slices of)
  ``flat_grad`` (parallel/ddp.py FlatGradReducer);
* ``zero_grad`` = one memset.

Semantics match ``torch.nn.utils.clip_grad_norm_`` +
``torch.optim.RMSprop`` (momentum=0; eps added AFTER sqrt), verified in
tests/test_optim.py. ``state_dict`` emits the torch-RMSprop-compatible
layout (per-param ``square_avg``), keeping the checkpoint format
contract.
"""
from __future__ import annotations

from typing import List

import torch

from torched_impala_amd import ops


class FlatRMSprop(torch.optim.Optimizer):
    def __init__(self, params, lr=6e-4, alpha=0.99, eps=0.01,
                 max_grad_norm=40.0, world_size=1):
        params = [p for p in params if p.requires_grad]
        assert params, "no parameters"
        super().__init__(params, dict(lr=lr, alpha=alpha, eps=eps))
        device = params[0].device
        dtype = params[0].dtype
        assert all(p.device == device and p.dtype == torch.float32
                   for p in params), "FlatRMSprop requires fp32 params on one device"
        del dtype
        self.max_grad_norm = float(max_grad_norm)
        # reverse order ~ backward completion order
        self._order: List[torch.nn.Parameter] = list(reversed(params))
        total = sum(p.numel() for p in self._order)
        self.flat_param = torch.zeros(total, device=device)
        self.flat_grad = torch.zeros(total, device=device)
        self.flat_sqavg = torch.zeros(total, device=device)
        self.slices = {}
        off = 0
        with torch.no_grad():
            for p in self._order:
                n = p.numel()
                self.flat_param[off:off + n].copy_(p.data.reshape(-1))
                p.data = self.flat_param[off:off + n].view_as(p)
                p.grad = self.flat_grad[off:off + n].view_as(p)
                self.slices[p] = (off, n)
                off += n
        self.lr_t = torch.zeros(1, device=device)
        self.norm_sq = torch.zeros(1, device=device)
        # non-finite-gradient skip counter (device; graph-capturable):
        # the fused kernel refuses to apply an update whose global norm
        # is inf/NaN — one poisoned gradient otherwise NaNs the params
        # permanently (0*inf=NaN even at lr=0)
        self.skip_t = torch.zeros(1, device=device)
        self.scale_t = torch.full((1,), 1.0 / world_size, device=device)
        self._steps = 0

    # --- torch Optimizer API ---
    def zero_grad(self, set_to_none: bool = False):  # noqa: ARG002
        self.flat_grad.zero_()
        # re-establish the grad views if something (e.g. module.zero_grad
        # with set_to_none=True) detached them
        for p in self._order:
            if p.grad is None or p.grad.data_ptr() != \
                    self.flat_grad[self.slices[p][0]].data_ptr():
                off, n = self.slices[p]
                p.grad = self.flat_grad[off:off + n].view_as(p)

    @torch.no_grad()
    def step(self, closure=None):  # noqa: ARG002
        self.sync_lr()
        self.step_device()
        return None

    # --- split API for graph capture ---
    def sync_lr(self):
        """Host-side: push the (scheduler-updated) python lr to device.
        Call OUTSIDE any captured graph, once per step."""
        self.lr_t.fill_(self.param_groups[0]["lr"])
        self._steps += 1
        # keep LRScheduler's step-order bookkeeping happy in graph mode,
        # where optimizer.step() itself is inside the captured graph
        self._step_count = getattr(self, "_step_count", 0) + 1
        self._opt_called = True  # torch>=2.4 scheduler-order check

    @torch.no_grad()
    def step_device(self):
        """Device-side update; every op here is graph-capturable."""
        self.norm_sq.zero_()
        g = self.param_groups[0]
        alpha, eps = g["alpha"], g["eps"]
        if self.flat_param.is_cuda and ops.hip_available():
            ops.ext_hip().fused_clip_rmsprop(
                self.flat_param, self.flat_grad, self.flat_sqavg,
                self.lr_t, self.norm_sq, self.scale_t,
                self.max_grad_norm, alpha, eps, self.skip_t)
            return
        # eager reference path (CPU tests / escape hatch) — same math
        gr = self.flat_grad * self.scale_t
        if self.max_grad_norm > 0:
            norm = gr.norm()
            if not torch.isfinite(norm):
                self.skip_t += 1  # skip the poisoned update (see above)
                return
            coef = torch.clamp(self.max_grad_norm / (norm + 1e-6), max=1.0)
            gr = gr * coef
        self.flat_sqavg.mul_(alpha).addcmul_(gr, gr, value=1.0 - alpha)
        denom = self.flat_sqavg.sqrt().add_(eps)
        self.flat_param.addcdiv_(gr, denom, value=-float(self.lr_t.item()))

    # --- checkpoint interop (torch.optim.RMSprop-compatible layout) ---
    def state_dict(self):
        params = self.param_groups[0]["params"]
        state = {}
        for i, p in enumerate(params):
            off, n = self.slices[p]
            state[i] = {
                "step": self._steps,
                "square_avg": self.flat_sqavg[off:off + n].view_as(p).clone(),
            }
        g = self.param_groups[0]
        groups = [{"lr": g["lr"], "alpha": g["alpha"], "eps": g["eps"],
                   "momentum": 0.0, "weight_decay": 0.0, "centered": False,
                   "params": list(range(len(params)))}]
        return {"state": state, "param_groups": groups}

    def load_state_dict(self, sd):
        params = self.param_groups[0]["params"]
        with torch.no_grad():
            for i, p in enumerate(params):
                if i in sd["state"] or str(i) in sd["state"]:
                    ent = sd["state"].get(i, sd["state"].get(str(i)))
                    off, n = self.slices[p]
                    self.flat_sqavg[off:off + n].copy_(
                        ent["square_avg"].reshape(-1).to(self.flat_sqavg.device))
                    self._steps = int(ent.get("step", self._steps))
        for k in ("lr", "alpha", "eps"):
            if sd.get("param_groups"):
                self.param_groups[0][k] = sd["param_groups"][0].get(
                    k, self.param_groups[0][k])
