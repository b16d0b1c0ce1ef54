"""Custom-op layer: gfx950 HIP kernels behind torch.autograd Functions.

Two in-tree extensions (built by ``__graft_entry__.build()`` /
``python setup.py build_ext`` -> ``torched_impala_amd/*.so``):

* ``_impala_ring`` — host-only C++: the lock-free MPMC queue.
* ``_impala_hip``  — gfx950 HIP kernels (V-trace scan, fused loss, LSTM
  cell, host pinning, async drain helpers).

Policy (per the round contract): on a GPU box the HIP path is the path —
if ``torch.cuda.is_available()`` and the extension is missing, ops FAIL
LOUDLY instead of silently falling back to eager. The eager composition
is used on CPU (tests) or when ``flags.use_hip_kernels=False``.
"""
from __future__ import annotations

import importlib
import logging
import os

import torch

log = logging.getLogger(__name__)

_EXT_CPU = None
_EXT_HIP = None


def ext_cpu():
    global _EXT_CPU
    if _EXT_CPU is None:
        _EXT_CPU = importlib.import_module("torched_impala_amd._impala_ring")
    return _EXT_CPU


def ext_hip(required: bool = True):
    global _EXT_HIP
    if _EXT_HIP is None:
        try:
            _EXT_HIP = importlib.import_module("torched_impala_amd._impala_hip")
        except ImportError:
            if required and torch.cuda.is_available() and not _eager_forced():
                raise RuntimeError(
                    "GPU present but torched_impala_amd._impala_hip is not "
                    "built — run __graft_entry__.build() (hipcc, gfx950). "
                    "Refusing silent eager fallback on a GPU box.")
            return None
    return _EXT_HIP


def _eager_forced() -> bool:
    return os.environ.get("IMPALA_AMD_EAGER", "0") == "1"


def hip_available() -> bool:
    """True iff we should dispatch to the HIP kernels."""
    if _eager_forced():
        return False
    if not torch.cuda.is_available():
        return False
    return ext_hip(required=True) is not None


def pin_host_memory(t: torch.Tensor):
    """hipHostRegister a CPU tensor's storage (page-lock for async DMA)."""
    ext = ext_hip(required=True)
    if ext is None:
        return
    ext.host_register(t)


def unpin_host_memory(t: torch.Tensor):
    ext = ext_hip(required=False)
    if ext is not None:
        ext.host_unregister(t)


# ---------------------------------------------------------------------------
# Fused V-trace + loss op
# ---------------------------------------------------------------------------

class _FusedVtraceLoss(torch.autograd.Function):
    """Forward: log-softmax/entropy row pass + V-trace reverse scan +
    loss reductions, in two HIP kernels. Backward: one elementwise kernel
    producing d(total_loss)/d(target_logits) and /d(values) exactly
    (vs/advantages are detached by construction, so no gradient flows
    through the scan — SURVEY.md §2e).
    """

    @staticmethod
    def forward(ctx, target_logits, values, bootstrap_value,
                behavior_logits, actions, rewards, discounts,
                clip_rho, clip_c, lambda_, clip_pg_rho,
                baseline_cost, entropy_cost):
        ext = ext_hip()
        T, B, A = target_logits.shape
        logp, target_lp, row_negent, vs, pg_adv = ext.vtrace_loss_fwd(
            target_logits.contiguous(), behavior_logits.contiguous(),
            actions.contiguous(), rewards.contiguous(),
            discounts.contiguous(), values.contiguous(),
            bootstrap_value.contiguous(),
            float(clip_rho), float(clip_c), float(lambda_),
            float(clip_pg_rho))
        pg = -(target_lp * pg_adv.flatten()).sum()
        vdiff = vs - values.detach()  # (T,B)
        bl = 0.5 * (vdiff * vdiff).sum()
        ent = row_negent.sum()  # sum of p*logp (negative entropy)
        loss = pg + baseline_cost * bl + entropy_cost * ent
        ctx.save_for_backward(logp, row_negent, actions, pg_adv, vdiff)
        ctx.coefs = (float(baseline_cost), float(entropy_cost))
        ctx.mark_non_differentiable(vs, pg_adv)
        return loss, pg.detach(), bl.detach(), ent.detach(), vs, pg_adv

    @staticmethod
    def backward(ctx, dloss, *unused):
        ext = ext_hip()
        logp, row_negent, actions, pg_adv, vdiff = ctx.saved_tensors
        baseline_cost, entropy_cost = ctx.coefs
        grad_scale = dloss.reshape(1).to(
            device=logp.device, dtype=torch.float32).contiguous()
        dlogits, dvalues = ext.vtrace_loss_bwd(
            logp, row_negent, actions, pg_adv.contiguous(),
            vdiff.contiguous(), grad_scale, baseline_cost, entropy_cost)
        return (dlogits, dvalues) + (None,) * 11


def fused_vtrace_loss(target_logits, values, bootstrap_value,
                      behavior_logits, actions, rewards, discounts,
                      clip_rho=1.0, clip_c=1.0, lambda_=1.0,
                      clip_pg_rho=None,
                      baseline_cost=0.5, entropy_cost=0.01,
                      force_eager: bool = False):
    """Returns (total_loss, stats dict, vs, pg_advantages).

    target_logits (T,B,A) and values (T,B) carry grad; everything else is
    data. Dispatches to the HIP fused path on GPU, eager elsewhere.
    clip_pg_rho (rho-bar for the PG advantage) defaults to clip_rho so
    the HIP and eager paths agree for any --clip-rho-threshold.
    """
    if clip_pg_rho is None:
        clip_pg_rho = clip_rho
    if not force_eager and hip_available() and target_logits.is_cuda:
        loss, pg, bl, ent, vs, adv = _FusedVtraceLoss.apply(
            target_logits, values, bootstrap_value, behavior_logits,
            actions, rewards, discounts, clip_rho, clip_c, lambda_,
            clip_pg_rho, baseline_cost, entropy_cost)
        stats = dict(pg_loss=pg.detach(), baseline_loss=bl.detach(),
                     entropy_loss=ent.detach())
        return loss, stats, vs, adv

    # eager composition (CPU tests / escape hatch)
    from torched_impala_amd.algo import vtrace as VT, losses as L
    vt = VT.from_logits(
        behavior_logits, target_logits, actions,
        discounts, rewards, values, bootstrap_value,
        clip_rho_threshold=clip_rho, clip_c_threshold=clip_c, lambda_=lambda_,
        clip_pg_rho_threshold=clip_pg_rho)
    loss, stats = L.total_loss(
        target_logits, actions, vt.pg_advantages, vt.vs, values,
        baseline_cost, entropy_cost)
    return loss, stats, vt.vs, vt.pg_advantages
