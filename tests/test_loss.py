"""Loss gradients: autograd on the eager composition vs hand-derived
closed forms (the formulas the fused HIP backward implements)."""
import torch
import torch.nn.functional as F

from torched_impala_amd import ops


def _setup(T=6, B=4, A=5, seed=0):
    torch.manual_seed(seed)
    target_logits = torch.randn(T, B, A, requires_grad=True)
    values = torch.randn(T, B, requires_grad=True)
    boot = torch.randn(B)
    behavior_logits = torch.randn(T, B, A)
    actions = torch.randint(0, A, (T, B))
    rewards = torch.randn(T, B)
    discounts = torch.full((T, B), 0.99)
    return target_logits, values, boot, behavior_logits, actions, rewards, discounts


def test_eager_loss_matches_closed_form_grads():
    tl, v, boot, bl, a, r, d = _setup()
    bc, ec = 0.5, 0.01
    loss, stats, vs, adv = ops.fused_vtrace_loss(
        tl, v, boot, bl, a, r, d, baseline_cost=bc, entropy_cost=ec,
        force_eager=True)
    loss.backward()

    # closed forms (what vtrace_loss_bwd computes):
    with torch.no_grad():
        p = F.softmax(tl, dim=-1)
        onehot = F.one_hot(a, p.shape[-1]).float()
        logp = F.log_softmax(tl, dim=-1)
        # pg: adv * (p - onehot)
        dlogits_pg = adv.unsqueeze(-1) * (p - onehot)
        # entropy loss = sum p*logp; d/dlogits = p * (logp - sum_k p_k logp_k)
        ent_inner = (p * logp).sum(-1, keepdim=True)
        dlogits_ent = p * (logp - ent_inner)
        dlogits = dlogits_pg + ec * dlogits_ent
        dvalues = -bc * (vs - v.detach())
    torch.testing.assert_close(tl.grad, dlogits, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(v.grad, dvalues, rtol=1e-4, atol=1e-5)


def test_loss_stats_finite_and_signs():
    tl, v, boot, bl, a, r, d = _setup(seed=3)
    loss, stats, vs, adv = ops.fused_vtrace_loss(
        tl, v, boot, bl, a, r, d, force_eager=True)
    assert torch.isfinite(loss)
    assert stats["baseline_loss"] >= 0
    # entropy_loss is negative entropy: in [-T*B*log(A), 0]
    assert stats["entropy_loss"] <= 0


def test_on_policy_zero_kl_entropy_value():
    """With uniform logits, negative entropy = -T*B*log(A)."""
    T, B, A = 4, 3, 7
    tl = torch.zeros(T, B, A, requires_grad=True)
    v = torch.zeros(T, B, requires_grad=True)
    loss, stats, _, _ = ops.fused_vtrace_loss(
        tl, v, torch.zeros(B), torch.zeros(T, B, A),
        torch.zeros(T, B, dtype=torch.long), torch.zeros(T, B),
        torch.full((T, B), 0.99), force_eager=True)
    expected = -T * B * torch.log(torch.tensor(float(A)))
    torch.testing.assert_close(stats["entropy_loss"], expected)


def test_finite_difference_gradients():
    """finite-difference check of the loss closed forms at a
    small size, with the V-trace targets HELD FIXED (the IMPALA loss is a
    semi-gradient: vs/advantages are detached by construction, so the
    total derivative through the scan must NOT appear in the gradient —
    finite differences are taken at frozen targets)."""
    from torched_impala_amd.algo import losses as L
    torch.manual_seed(11)
    T, B, A = 3, 2, 3
    tl = torch.randn(T, B, A, dtype=torch.float64, requires_grad=True)
    v = torch.randn(T, B, dtype=torch.float64, requires_grad=True)
    a = torch.randint(0, A, (T, B))
    adv = torch.randn(T, B, dtype=torch.float64)   # frozen targets
    vs = torch.randn(T, B, dtype=torch.float64)

    def f(tl_, v_):
        loss, _ = L.total_loss(tl_, a, adv, vs, v_, 0.5, 0.01)
        return loss

    loss = f(tl, v)
    loss.backward()
    eps = 1e-6
    for idx in [(0, 0, 0), (1, 1, 2), (2, 0, 1)]:
        with torch.no_grad():
            tp = tl.detach().clone(); tp[idx] += eps
            tm = tl.detach().clone(); tm[idx] -= eps
        num = (f(tp, v.detach()) - f(tm, v.detach())) / (2 * eps)
        assert abs(float(num) - float(tl.grad[idx])) < 1e-5, \
            f"dlogits{idx}: fd={float(num)} autograd={float(tl.grad[idx])}"
    for idx in [(0, 0), (2, 1)]:
        with torch.no_grad():
            vp = v.detach().clone(); vp[idx] += eps
            vm = v.detach().clone(); vm[idx] -= eps
        num = (f(tl.detach(), vp) - f(tl.detach(), vm)) / (2 * eps)
        assert abs(float(num) - float(v.grad[idx])) < 1e-5, \
            f"dvalues{idx}: fd={float(num)} autograd={float(v.grad[idx])}"
