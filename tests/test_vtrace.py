"""V-trace correctness (SURVEY.md §4.1 — the single most important test):
the O(T) reverse-scan implementation vs an O(T^2) naive oracle, and the
on-policy reduction to n-step Bellman targets."""
import numpy as np
import pytest
import torch

from torched_impala_amd.algo import vtrace


def naive_vtrace(log_rhos, discounts, rewards, values, bootstrap,
                 clip_rho=1.0, clip_c=1.0, lam=1.0):
    """O(T^2) double-loop ground truth, straight from the paper's Eq. (1)."""
    T, B = rewards.shape
    rhos = np.exp(log_rhos)
    crho = np.minimum(rhos, clip_rho)
    cs = lam * np.minimum(rhos, clip_c)
    values_tp1 = np.concatenate([values[1:], bootstrap[None]], axis=0)
    deltas = crho * (rewards + discounts * values_tp1 - values)
    vs = np.zeros((T, B))
    for s in range(T):
        acc = values[s].copy()
        for t in range(s, T):
            prod_c = np.ones(B)
            for i in range(s, t):
                prod_c *= discounts[i] * cs[i]
            acc += prod_c * deltas[t]
        vs[s] = acc
    vs_tp1 = np.concatenate([vs[1:], bootstrap[None]], axis=0)
    pg_adv = np.minimum(rhos, clip_rho) * (rewards + discounts * vs_tp1 - values)
    return vs, pg_adv


def _rand_inputs(T=10, B=5, seed=0):
    rng = np.random.default_rng(seed)
    log_rhos = rng.normal(0, 0.6, (T, B))
    discounts = (rng.random((T, B)) > 0.1) * 0.99
    rewards = rng.normal(0, 1, (T, B))
    values = rng.normal(0, 1, (T + 1, B))
    return log_rhos, discounts, rewards, values[:-1], values[-1]


@pytest.mark.parametrize("seed", [0, 1, 2])
def test_scan_matches_naive(seed):
    log_rhos, discounts, rewards, values, boot = _rand_inputs(seed=seed)
    vs_ref, adv_ref = naive_vtrace(log_rhos, discounts, rewards, values, boot)
    out = vtrace.from_importance_weights(
        torch.tensor(log_rhos), torch.tensor(discounts),
        torch.tensor(rewards), torch.tensor(values), torch.tensor(boot))
    np.testing.assert_allclose(out.vs.numpy(), vs_ref, rtol=1e-6, atol=1e-6)
    np.testing.assert_allclose(out.pg_advantages.numpy(), adv_ref,
                               rtol=1e-6, atol=1e-6)


def test_on_policy_reduces_to_nstep_bellman():
    """pi == mu, clip thresholds >= 1 => vs_s = n-step return."""
    T, B = 8, 3
    rng = np.random.default_rng(42)
    discounts = np.full((T, B), 0.9)
    rewards = rng.normal(0, 1, (T, B))
    values = rng.normal(0, 1, (T + 1, B))
    log_rhos = np.zeros((T, B))  # on-policy
    out = vtrace.from_importance_weights(
        torch.tensor(log_rhos), torch.tensor(discounts),
        torch.tensor(rewards), torch.tensor(values[:-1]),
        torch.tensor(values[-1]))
    # n-step Bellman target: sum_{t=s}^{T-1} gamma^{t-s} r_t + gamma^{T-s} V_T
    for s in range(T):
        target = np.zeros(B)
        g = np.ones(B)
        for t in range(s, T):
            target += g * rewards[t]
            g *= discounts[t]
        target += g * values[T]
        np.testing.assert_allclose(out.vs[s].numpy(), target, rtol=1e-5,
                                   atol=1e-5)


def test_from_logits_consistency():
    """from_logits == from_importance_weights on its own log-rhos."""
    T, B, A = 6, 4, 5
    torch.manual_seed(0)
    behavior = torch.randn(T, B, A)
    target = torch.randn(T, B, A)
    actions = torch.randint(0, A, (T, B))
    discounts = torch.full((T, B), 0.99)
    rewards = torch.randn(T, B)
    values = torch.randn(T, B)
    boot = torch.randn(B)
    out = vtrace.from_logits(behavior, target, actions, discounts, rewards,
                             values, boot)
    tlp = vtrace.action_log_probs(target, actions)
    blp = vtrace.action_log_probs(behavior, actions)
    ref = vtrace.from_importance_weights(tlp - blp, discounts, rewards,
                                         values, boot)
    torch.testing.assert_close(out.vs, ref.vs)
    torch.testing.assert_close(out.pg_advantages, ref.pg_advantages)
    # log-prob sanity: gather of log_softmax
    lsm = torch.log_softmax(target, dim=-1)
    ref_lp = lsm.gather(2, actions.unsqueeze(-1)).squeeze(-1)
    torch.testing.assert_close(out.target_action_log_probs, ref_lp)


@pytest.mark.parametrize("clip_rho,clip_c,lam", [(0.8, 1.2, 1.0),
                                                 (1.5, 0.7, 0.95),
                                                 (2.0, 2.0, 0.5)])
def test_scan_matches_naive_nondefault_clips(clip_rho, clip_c, lam):
    log_rhos, discounts, rewards, values, boot = _rand_inputs(seed=9)
    vs_ref, adv_ref = naive_vtrace(log_rhos, discounts, rewards, values,
                                   boot, clip_rho=clip_rho, clip_c=clip_c,
                                   lam=lam)
    out = vtrace.from_importance_weights(
        torch.tensor(log_rhos), torch.tensor(discounts),
        torch.tensor(rewards), torch.tensor(values), torch.tensor(boot),
        clip_rho_threshold=clip_rho, clip_c_threshold=clip_c, lambda_=lam,
        clip_pg_rho_threshold=clip_rho)
    np.testing.assert_allclose(out.vs.numpy(), vs_ref, rtol=1e-6, atol=1e-6)
    np.testing.assert_allclose(out.pg_advantages.numpy(), adv_ref,
                               rtol=1e-6, atol=1e-6)
