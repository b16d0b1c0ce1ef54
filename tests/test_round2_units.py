"""Unit coverage for round-2 utilities (CPU-safe)."""
import os

import pytest
import torch

from torched_impala_amd.actor import resolve_actor_tier


def test_resolve_actor_tier_passthrough():
    for t in ("off", "safe", "frozen", "native"):
        assert resolve_actor_tier(t) == t


def test_resolve_actor_tier_bf16_resolves_to_real_tier():
    got = resolve_actor_tier("native-bf16")
    # on any host this must resolve to an ACHIEVABLE tier, never stay
    # a wish: bf16 only when the ISA probe says so
    assert got in ("native-bf16", "native", "frozen")


def test_resolve_actor_tier_non_atari_obs():
    assert resolve_actor_tier("native-bf16", obs_shape=(3, 72, 96)) == "frozen"


def test_shallow_mfma_gate_env(monkeypatch):
    from torched_impala_amd.ops import shallow_mfma as S
    monkeypatch.delenv("IMPALA_SHALLOW_MFMA", raising=False)
    assert S.shallow_mfma_enabled()
    monkeypatch.setenv("IMPALA_SHALLOW_MFMA", "0")
    assert not S.shallow_mfma_enabled()


def test_lstm_bf16_gate_default_off(monkeypatch):
    from torched_impala_amd.ops import lstm as L
    monkeypatch.delenv("IMPALA_LSTM_BF16", raising=False)
    assert not L.lstm_bf16_weights()
    monkeypatch.setenv("IMPALA_LSTM_BF16", "1")
    assert L.lstm_bf16_weights()


def test_graphed_step_destroy_idempotent():
    from torched_impala_amd import learner as L
    from torched_impala_amd.config import Flags
    flags = Flags(env="synthetic-atari")
    g = L.GraphedStep(flags, model=None, optimizer=None, batcher=None)
    # no graphs captured: destroy must be a no-op (no CUDA calls)
    g.destroy()
    g.destroy()
    assert g.graphs == {} and g.outputs == {}


def test_mirror_poll_without_pending_is_noop():
    from torched_impala_amd.models import MLPPolicy
    from torched_impala_amd.transport.weights import WeightMirror
    m = MLPPolicy((4,), 2)
    mir = WeightMirror(m)
    v0 = int(mir.version.item())
    mir.poll_publish()
    assert int(mir.version.item()) == v0


def test_linear_relu_cpu_fallback_matches_eager():
    from torched_impala_amd.ops.linear import linear_relu
    torch.manual_seed(0)
    fc = torch.nn.Linear(8, 5)
    x = torch.randn(3, 8)
    got = linear_relu(fc, x)
    want = torch.relu(fc(x))
    torch.testing.assert_close(got, want, rtol=0, atol=0)
