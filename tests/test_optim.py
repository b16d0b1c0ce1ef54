"""FlatRMSprop parity vs torch clip_grad_norm_ + torch.optim.RMSprop
(the semantics contract; the HIP kernels are checked against the same
reference in tests/test_gpu_kernels.py)."""
import torch
import pytest

from torched_impala_amd.models import MLPPolicy
from torched_impala_amd.ops.optim import FlatRMSprop


def _loss(model):
    x = dict(frame=torch.randn(3, 4, 4), reward=torch.zeros(3, 4),
             done=torch.zeros(3, 4, dtype=torch.bool),
             last_action=torch.zeros(3, 4, dtype=torch.long))
    out, _ = model(x, ())
    return out["policy_logits"].pow(2).sum() + 3 * out["baseline"].sum()


@pytest.mark.parametrize("max_norm", [40.0, 0.05])
def test_flat_rmsprop_matches_torch(max_norm):
    torch.manual_seed(0)
    m1 = MLPPolicy((4,), 2)
    m2 = MLPPolicy((4,), 2)
    m2.load_state_dict(m1.state_dict())

    opt1 = FlatRMSprop(m1.parameters(), lr=0.01, alpha=0.99, eps=0.01,
                       max_grad_norm=max_norm)
    opt2 = torch.optim.RMSprop(m2.parameters(), lr=0.01, alpha=0.99, eps=0.01)

    for it in range(5):
        torch.manual_seed(100 + it)
        opt1.zero_grad()
        _loss(m1).backward()
        opt1.step()

        torch.manual_seed(100 + it)
        opt2.zero_grad()
        _loss(m2).backward()
        torch.nn.utils.clip_grad_norm_(m2.parameters(), max_norm)
        opt2.step()

    for (k1, p1), (k2, p2) in zip(m1.state_dict().items(),
                                  m2.state_dict().items()):
        assert k1 == k2
        torch.testing.assert_close(p1, p2, rtol=1e-5, atol=1e-6,
                                   msg=lambda m: f"{k1}: {m}")


def test_flat_rmsprop_state_dict_roundtrip():
    torch.manual_seed(1)
    m = MLPPolicy((4,), 2)
    opt = FlatRMSprop(m.parameters(), lr=0.01)
    for it in range(3):
        opt.zero_grad()
        _loss(m).backward()
        opt.step()
    sd = opt.state_dict()
    assert "state" in sd and "param_groups" in sd
    assert sd["state"][0]["square_avg"].shape == \
        list(m.parameters())[0].shape

    m2 = MLPPolicy((4,), 2)
    m2.load_state_dict(m.state_dict())
    opt2 = FlatRMSprop(m2.parameters(), lr=0.01)
    opt2.load_state_dict(sd)
    torch.testing.assert_close(opt2.flat_sqavg, opt.flat_sqavg)

    # interop: torch RMSprop can load it too
    opt3 = torch.optim.RMSprop(m2.parameters(), lr=0.01, alpha=0.99, eps=0.01)
    opt3.load_state_dict(sd)


def test_flat_rmsprop_grad_views():
    """p.grad views alias the flat buffer; zero_grad is one memset."""
    m = MLPPolicy((4,), 2)
    opt = FlatRMSprop(m.parameters(), lr=0.01)
    _loss(m).backward()
    assert opt.flat_grad.abs().sum() > 0
    for p in m.parameters():
        assert p.grad.data_ptr() >= opt.flat_grad.data_ptr()
    opt.zero_grad()
    assert opt.flat_grad.abs().sum() == 0
    for p in m.parameters():
        assert p.grad.abs().sum() == 0


def test_nonfinite_grad_skips_update():
    """One inf/NaN gradient element must SKIP the update entirely
    (params and sqavg untouched, skip counter bumped) — a poisoned
    gradient otherwise NaNs the parameters permanently (0*inf=NaN even
    at lr=0)."""
    import torch
    from torched_impala_amd.ops.optim import FlatRMSprop
    p = torch.nn.Parameter(torch.randn(64))
    opt = FlatRMSprop([p], lr=0.01, max_grad_norm=40.0)
    p0 = opt.flat_param.clone()
    sq0 = opt.flat_sqavg.clone()
    opt.flat_grad.normal_()
    opt.flat_grad[7] = float("inf")
    opt.step()
    assert float(opt.skip_t) == 1.0
    torch.testing.assert_close(opt.flat_param, p0, rtol=0, atol=0)
    torch.testing.assert_close(opt.flat_sqavg, sq0, rtol=0, atol=0)
    # a clean grad afterwards applies normally
    opt.flat_grad.normal_()
    opt.step()
    assert float(opt.skip_t) == 1.0
    assert not torch.equal(opt.flat_param, p0)
