"""V-trace importance-weighted return/advantage targets.

The algorithmic core of IMPALA (Espeholt et al. 2018, §4.1; restated in
SURVEY.md §2e).  This module is the pure-PyTorch reference; the gfx950
HIP kernel path (torched_impala_amd.ops) is validated against it, and it
itself is validated against an O(T^2) naive oracle in tests/test_vtrace.py.

All tensors are time-major: (T, B) or (T, B, A).
"""
from __future__ import annotations

import collections

import torch
import torch.nn.functional as F

VTraceReturns = collections.namedtuple("VTraceReturns", ["vs", "pg_advantages"])
VTraceFromLogitsReturns = collections.namedtuple(
    "VTraceFromLogitsReturns",
    ["vs", "pg_advantages", "log_rhos", "behavior_action_log_probs",
     "target_action_log_probs"])


def action_log_probs(policy_logits: torch.Tensor, actions: torch.Tensor):
    """log pi(a|x) for time-major (T, B, A) logits, (T, B) actions."""
    return -F.nll_loss(
        F.log_softmax(policy_logits.flatten(0, 1), dim=-1),
        actions.flatten(0, 1).long(),
        reduction="none").view_as(actions).float()


@torch.no_grad()
def from_importance_weights(
    log_rhos: torch.Tensor,       # (T, B)
    discounts: torch.Tensor,      # (T, B)  gamma * (1 - done_{t+1})
    rewards: torch.Tensor,        # (T, B)
    values: torch.Tensor,         # (T, B)  V(x_t) from target policy
    bootstrap_value: torch.Tensor,  # (B,)  V(x_{T})
    clip_rho_threshold: float = 1.0,
    clip_c_threshold: float = 1.0,
    lambda_: float = 1.0,
    clip_pg_rho_threshold: float = 1.0,
) -> VTraceReturns:
    """The reverse scan: vs_t = V(x_t) + A_t,
    A_t = delta_t + discount_t * c_t * A_{t+1}."""
    rhos = torch.exp(log_rhos)
    clipped_rhos = torch.clamp(rhos, max=clip_rho_threshold)
    cs = lambda_ * torch.clamp(rhos, max=clip_c_threshold)

    values_t_plus_1 = torch.cat(
        [values[1:], bootstrap_value.unsqueeze(0)], dim=0)
    deltas = clipped_rhos * (rewards + discounts * values_t_plus_1 - values)

    T = deltas.shape[0]
    acc = torch.zeros_like(bootstrap_value)
    accs = [None] * T
    for t in reversed(range(T)):
        acc = deltas[t] + discounts[t] * cs[t] * acc
        accs[t] = acc
    vs_minus_v = torch.stack(accs, dim=0)
    vs = vs_minus_v + values

    vs_t_plus_1 = torch.cat([vs[1:], bootstrap_value.unsqueeze(0)], dim=0)
    clipped_pg_rhos = torch.clamp(rhos, max=clip_pg_rho_threshold)
    pg_advantages = clipped_pg_rhos * (rewards + discounts * vs_t_plus_1 - values)
    return VTraceReturns(vs=vs, pg_advantages=pg_advantages)


def from_logits(
    behavior_policy_logits: torch.Tensor,  # (T, B, A), recorded by the actor
    target_policy_logits: torch.Tensor,    # (T, B, A), learner forward
    actions: torch.Tensor,                 # (T, B)
    discounts: torch.Tensor,               # (T, B)
    rewards: torch.Tensor,                 # (T, B)
    values: torch.Tensor,                  # (T, B)
    bootstrap_value: torch.Tensor,         # (B,)
    clip_rho_threshold: float = 1.0,
    clip_c_threshold: float = 1.0,
    lambda_: float = 1.0,
    clip_pg_rho_threshold: float = None,
) -> VTraceFromLogitsReturns:
    if clip_pg_rho_threshold is None:
        clip_pg_rho_threshold = clip_rho_threshold
    target_lp = action_log_probs(target_policy_logits, actions)
    behavior_lp = action_log_probs(behavior_policy_logits, actions)
    log_rhos = (target_lp - behavior_lp).detach()
    vt = from_importance_weights(
        log_rhos=log_rhos,
        discounts=discounts, rewards=rewards,
        values=values.detach(), bootstrap_value=bootstrap_value.detach(),
        clip_rho_threshold=clip_rho_threshold,
        clip_c_threshold=clip_c_threshold,
        lambda_=lambda_,
        clip_pg_rho_threshold=clip_pg_rho_threshold)
    return VTraceFromLogitsReturns(
        vs=vt.vs, pg_advantages=vt.pg_advantages, log_rhos=log_rhos,
        behavior_action_log_probs=behavior_lp,
        target_action_log_probs=target_lp)
