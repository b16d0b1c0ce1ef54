"""HIP kernel parity vs the pure-PyTorch fp32 reference.
All tests need a real MI355X — marked gpu."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from torched_impala_amd import ops
from torched_impala_amd.algo import vtrace


def _dev():
    return torch.device("cuda", 0)


def test_native_extension_loaded():
    assert torch.cuda.is_available()
    assert ops.hip_available(), "HIP extension must load on a GPU box"
    ext = ops.ext_hip()
    assert hasattr(ext, "vtrace_scan")


@pytest.mark.parametrize("T,B", [(20, 32), (100, 512), (7, 3), (150, 64)])
def test_vtrace_scan_parity(T, B):
    torch.manual_seed(0)
    dev = _dev()
    log_rhos = (0.6 * torch.randn(T, B)).to(dev)
    discounts = (torch.rand(T, B, device=dev) > 0.1).float() * 0.99
    rewards = torch.randn(T, B, device=dev)
    values = torch.randn(T, B, device=dev)
    boot = torch.randn(B, device=dev)
    vs, adv = ops.ext_hip().vtrace_scan(
        log_rhos, discounts, rewards, values, boot, 1.0, 1.0, 1.0)
    ref = vtrace.from_importance_weights(
        log_rhos.cpu(), discounts.cpu(), rewards.cpu(), values.cpu(),
        boot.cpu())
    torch.testing.assert_close(vs.cpu(), ref.vs, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(adv.cpu(), ref.pg_advantages, rtol=1e-5,
                               atol=1e-5)


@pytest.mark.parametrize("T,B,A", [(20, 32, 6), (10, 8, 18), (100, 128, 15)])
def test_fused_loss_parity_and_grads(T, B, A):
    torch.manual_seed(1)
    dev = _dev()
    tl = torch.randn(T, B, A, device=dev, requires_grad=True)
    v = torch.randn(T, B, device=dev, requires_grad=True)
    boot = torch.randn(B, device=dev)
    bl = torch.randn(T, B, A, device=dev)
    a = torch.randint(0, A, (T, B), device=dev)
    r = torch.randn(T, B, device=dev)
    d = torch.full((T, B), 0.99, device=dev)

    loss, stats, vs, adv = ops.fused_vtrace_loss(tl, v, boot, bl, a, r, d)
    loss.backward()
    g_tl, g_v = tl.grad.clone(), v.grad.clone()

    tl2 = tl.detach().clone().requires_grad_()
    v2 = v.detach().clone().requires_grad_()
    loss_e, stats_e, vs_e, adv_e = ops.fused_vtrace_loss(
        tl2, v2, boot, bl, a, r, d, force_eager=True)
    loss_e.backward()

    torch.testing.assert_close(loss, loss_e, rtol=1e-4, atol=1e-4)
    for k in stats:
        torch.testing.assert_close(stats[k], stats_e[k], rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(vs, vs_e, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(adv, adv_e, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(g_tl, tl2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(g_v, v2.grad, rtol=1e-4, atol=1e-5)


def test_fused_loss_grad_scale():
    """backward with a non-unit upstream gradient scales grads linearly."""
    torch.manual_seed(2)
    dev = _dev()
    T, B, A = 6, 4, 5
    boot = torch.randn(B, device=dev)
    bl = torch.randn(T, B, A, device=dev)
    a = torch.randint(0, A, (T, B), device=dev)
    r = torch.randn(T, B, device=dev)
    d = torch.full((T, B), 0.99, device=dev)
    tl = torch.randn(T, B, A, device=dev, requires_grad=True)
    v = torch.randn(T, B, device=dev, requires_grad=True)

    loss, *_ = ops.fused_vtrace_loss(tl, v, boot, bl, a, r, d)
    (2.5 * loss).backward()

    tl2 = tl.detach().clone().requires_grad_()
    v2 = v.detach().clone().requires_grad_()
    loss2, *_ = ops.fused_vtrace_loss(tl2, v2, boot, bl, a, r, d)
    loss2.backward()

    torch.testing.assert_close(tl.grad, 2.5 * tl2.grad, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(v.grad, 2.5 * v2.grad, rtol=1e-5, atol=1e-6)


def test_pin_host_memory_roundtrip():
    t = torch.arange(1 << 20, dtype=torch.float32)
    t.share_memory_()
    ops.pin_host_memory(t)
    d = torch.empty_like(t, device=_dev())
    s = torch.cuda.Stream()
    with torch.cuda.stream(s):
        d.copy_(t, non_blocking=True)
    s.synchronize()
    assert torch.equal(d.cpu(), t)
    ops.unpin_host_memory(t)


def test_scan_handles_done_boundaries():
    """discount=0 rows cut the recursion exactly like the reference."""
    torch.manual_seed(3)
    dev = _dev()
    T, B = 20, 32
    log_rhos = 0.3 * torch.randn(T, B, device=dev)
    discounts = torch.full((T, B), 0.99, device=dev)
    discounts[5], discounts[13] = 0.0, 0.0
    rewards = torch.randn(T, B, device=dev)
    values = torch.randn(T, B, device=dev)
    boot = torch.randn(B, device=dev)
    vs, adv = ops.ext_hip().vtrace_scan(
        log_rhos, discounts, rewards, values, boot, 1.0, 1.0, 1.0)
    ref = vtrace.from_importance_weights(
        log_rhos.cpu(), discounts.cpu(), rewards.cpu(), values.cpu(), boot.cpu())
    torch.testing.assert_close(vs.cpu(), ref.vs, rtol=1e-5, atol=1e-5)
