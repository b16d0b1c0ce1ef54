"""IMPALA losses (paper canonical coefficients).

L = pg + baseline_cost * baseline + entropy_cost * entropy, where

  pg        = -sum_t  log pi(a_t|x_t) * pg_advantage_t         (adv detached)
  baseline  = 0.5 * sum_t (vs_t - V(x_t))^2                    (vs detached)
  entropy   = -sum_t H(pi(.|x_t))   (i.e. the NEGATIVE entropy, minimized)

The eager composition below is the reference; the fused HIP kernel
(ops.fused_vtrace_loss) computes the same three scalars and their exact
gradients in one pass.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F


def _fp(x: torch.Tensor) -> torch.Tensor:
    """Upcast low-precision logits to fp32; keep fp64 as-is (tests use
    fp64 for finite-difference checks)."""
    return x if x.dtype == torch.float64 else x.float()


def pg_loss(target_logits, actions, advantages):
    logp = F.log_softmax(_fp(target_logits.flatten(0, 1)), dim=-1)
    lp_a = logp.gather(
        1, actions.flatten(0, 1).long().unsqueeze(1)).squeeze(1)
    return -(lp_a * advantages.flatten(0, 1).detach()).sum()


def baseline_loss(vs, values):
    return 0.5 * ((vs.detach() - values) ** 2).sum()


def entropy_loss(target_logits):
    """Negative entropy summed over (T, B) — added with positive coef."""
    logits = _fp(target_logits.flatten(0, 1))
    logp = F.log_softmax(logits, dim=-1)
    p = F.softmax(logits, dim=-1)
    return (p * logp).sum()


def total_loss(target_logits, actions, advantages, vs, values,
               baseline_cost: float, entropy_cost: float):
    pg = pg_loss(target_logits, actions, advantages)
    bl = baseline_loss(vs, values)
    ent = entropy_loss(target_logits)
    return pg + baseline_cost * bl + entropy_cost * ent, dict(
        pg_loss=pg.detach(), baseline_loss=bl.detach(), entropy_loss=ent.detach())
