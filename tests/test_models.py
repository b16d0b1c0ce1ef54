import torch
import pytest

from torched_impala_amd.models import MLPPolicy, ShallowNet, DeepResNet


def _inputs(T, B, obs_shape, A, dtype=torch.uint8):
    if dtype == torch.uint8:
        frame = torch.randint(0, 256, (T, B, *obs_shape), dtype=dtype)
    else:
        frame = torch.randn(T, B, *obs_shape)
    return dict(frame=frame,
                reward=torch.randn(T, B),
                done=torch.zeros(T, B, dtype=torch.bool),
                last_action=torch.randint(0, A, (T, B)))


@pytest.mark.parametrize("use_lstm", [False, True])
def test_mlp_shapes(use_lstm):
    T, B, A = 5, 3, 2
    net = MLPPolicy((4,), A, use_lstm=use_lstm)
    inp = _inputs(T, B, (4,), A, dtype=torch.float32)
    out, core = net(inp, net.initial_state(B))
    assert out["policy_logits"].shape == (T, B, A)
    assert out["baseline"].shape == (T, B)
    assert out["action"].shape == (T, B)
    assert len(core) == (2 if use_lstm else 0)


@pytest.mark.parametrize("use_lstm", [False, True])
def test_shallow_shapes(use_lstm):
    T, B, A = 3, 2, 6
    net = ShallowNet((4, 84, 84), A, use_lstm=use_lstm)
    assert net.flat == 32 * 7 * 7
    out, _ = net(_inputs(T, B, (4, 84, 84), A), net.initial_state(B))
    assert out["policy_logits"].shape == (T, B, A)


def test_deep_shapes():
    T, B, A = 2, 2, 15
    net = DeepResNet((3, 72, 96), A, use_lstm=True)
    assert net.flat == 32 * 9 * 12
    out, core = net(_inputs(T, B, (3, 72, 96), A), net.initial_state(B))
    assert out["policy_logits"].shape == (T, B, A)
    assert core[0].shape == (B, 256)


def test_lstm_done_resets_state():
    """A done flag at step t must reset the carried state: outputs after
    the reset must equal a fresh-state run."""
    torch.manual_seed(0)
    T, B, A = 6, 1, 4
    net = MLPPolicy((4,), A, use_lstm=True)
    net.eval()
    inp = _inputs(T, B, (4,), A, dtype=torch.float32)
    inp["done"][3, 0] = True  # reset before processing step 3
    out1, _ = net(inp, net.initial_state(B))
    tail = {k: v[3:] for k, v in inp.items()}
    out2, _ = net(tail, net.initial_state(B))
    torch.testing.assert_close(out1["policy_logits"][3:], out2["policy_logits"])


def test_greedy_vs_sample():
    torch.manual_seed(0)
    net = MLPPolicy((4,), 2)
    inp = _inputs(1, 8, (4,), 2, dtype=torch.float32)
    out, _ = net(inp, (), greedy=True)
    assert (out["action"] == out["policy_logits"].argmax(-1)).all()


def test_checkpoint_param_names_frozen():
    """The frozen checkpoint naming contract."""
    net = ShallowNet((4, 84, 84), 6, use_lstm=True)
    names = set(net.state_dict().keys())
    for expect in ["conv1.weight", "conv2.weight", "conv3.weight",
                   "fc.weight", "core.weight_ih", "core.weight_hh",
                   "policy.weight", "baseline.weight"]:
        assert expect in names, f"missing frozen param name {expect}"
