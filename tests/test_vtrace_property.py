"""Property-based V-trace tests (hypothesis): the O(T) scan must match
the O(T^2) paper-equation oracle for ARBITRARY shapes, clip levels,
lambda, done patterns and importance-ratio magnitudes — not just the
fixed seeds in test_vtrace.py."""
import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from torched_impala_amd.algo import vtrace
from tests.test_vtrace import naive_vtrace


@settings(max_examples=40, deadline=None)
@given(
    T=st.integers(1, 24),
    B=st.integers(1, 7),
    clip_rho=st.floats(0.1, 4.0),
    clip_c=st.floats(0.1, 4.0),
    lam=st.floats(0.0, 1.0),
    rho_scale=st.floats(0.0, 2.5),
    done_p=st.floats(0.0, 0.9),
    seed=st.integers(0, 2**31 - 1),
)
def test_scan_matches_oracle_property(T, B, clip_rho, clip_c, lam,
                                      rho_scale, done_p, seed):
    rng = np.random.default_rng(seed)
    log_rhos = rng.normal(0, rho_scale, (T, B))
    discounts = (rng.random((T, B)) > done_p) * 0.99
    rewards = rng.normal(0, 1, (T, B))
    values = rng.normal(0, 1, (T + 1, B))

    want_vs, want_adv = naive_vtrace(
        log_rhos, discounts, rewards, values[:-1], values[-1],
        clip_rho=clip_rho, clip_c=clip_c, lam=lam)
    out = vtrace.from_importance_weights(
        log_rhos=torch.from_numpy(log_rhos).float(),
        discounts=torch.from_numpy(discounts).float(),
        rewards=torch.from_numpy(rewards).float(),
        values=torch.from_numpy(values[:-1]).float(),
        bootstrap_value=torch.from_numpy(values[-1]).float(),
        clip_rho_threshold=clip_rho, clip_pg_rho_threshold=clip_rho,
        clip_c_threshold=clip_c, lambda_=lam)
    np.testing.assert_allclose(out.vs.numpy(), want_vs, rtol=2e-4,
                               atol=2e-4)
    np.testing.assert_allclose(out.pg_advantages.numpy(), want_adv,
                               rtol=2e-4, atol=2e-4)


@settings(max_examples=25, deadline=None)
@given(T=st.integers(1, 16), B=st.integers(1, 5),
       seed=st.integers(0, 2**31 - 1))
def test_terminal_rows_isolate_episodes(T, B, seed):
    """A zero discount at row t makes every vs[s<=t] independent of all
    inputs after t — episode boundaries must not leak."""
    rng = np.random.default_rng(seed)
    log_rhos = rng.normal(0, 0.5, (T, B))
    discounts = np.full((T, B), 0.99)
    cut = rng.integers(0, T)
    discounts[cut] = 0.0
    rewards = rng.normal(0, 1, (T, B))
    values = rng.normal(0, 1, (T + 1, B))

    def run(rew, val):
        return vtrace.from_importance_weights(
            log_rhos=torch.from_numpy(log_rhos).float(),
            discounts=torch.from_numpy(discounts).float(),
            rewards=torch.from_numpy(rew).float(),
            values=torch.from_numpy(val[:-1]).float(),
            bootstrap_value=torch.from_numpy(val[-1]).float()).vs.numpy()

    base = run(rewards, values)
    rewards2 = rewards.copy()
    values2 = values.copy()
    if cut + 1 < T:
        rewards2[cut + 1:] += rng.normal(0, 5, (T - cut - 1, B))
        values2[cut + 2:] += rng.normal(0, 5, (T - cut - 1, B))
    else:
        values2[-1] += 7.0  # only the bootstrap row changes
    pert = run(rewards2, values2)
    np.testing.assert_allclose(base[:cut + 1], pert[:cut + 1], rtol=1e-5,
                               atol=1e-5)
