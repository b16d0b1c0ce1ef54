"""Property-based loss tests (hypothesis): closed-form gradient
identities must hold for arbitrary shapes, scales and coefficients —
the same identities the fused HIP backward implements
(csrc/vtrace_kernels.hip loss bwd)."""
import torch
import torch.nn.functional as F
from hypothesis import given, settings, strategies as st

from torched_impala_amd.algo import losses


@settings(max_examples=30, deadline=None)
@given(T=st.integers(1, 12), B=st.integers(1, 6), A=st.integers(2, 18),
       scale=st.floats(0.1, 8.0), seed=st.integers(0, 2**31 - 1))
def test_pg_grad_identity(T, B, A, scale, seed):
    """d pg / d logits = (softmax - onehot(a)) * adv per row."""
    g = torch.Generator().manual_seed(seed)
    logits = (torch.randn(T, B, A, generator=g) * scale).requires_grad_(True)
    actions = torch.randint(0, A, (T, B), generator=g)
    adv = torch.randn(T, B, generator=g)
    losses.pg_loss(logits, actions, adv).backward()
    p = F.softmax(logits.detach().flatten(0, 1), dim=-1)
    oh = F.one_hot(actions.flatten(), A).float()
    want = ((p - oh) * adv.flatten().unsqueeze(1)).view(T, B, A)
    torch.testing.assert_close(logits.grad, want, rtol=1e-4, atol=1e-5)


@settings(max_examples=30, deadline=None)
@given(T=st.integers(1, 12), B=st.integers(1, 6), A=st.integers(2, 18),
       scale=st.floats(0.1, 8.0), seed=st.integers(0, 2**31 - 1))
def test_entropy_grad_identity(T, B, A, scale, seed):
    """d(-H)/d logits = p * (logp - sum_j p_j logp_j) per row."""
    g = torch.Generator().manual_seed(seed)
    logits = (torch.randn(T, B, A, generator=g) * scale).requires_grad_(True)
    losses.entropy_loss(logits).backward()
    flat = logits.detach().flatten(0, 1)
    p = F.softmax(flat, dim=-1)
    lp = F.log_softmax(flat, dim=-1)
    want = (p * (lp - (p * lp).sum(-1, keepdim=True))).view(T, B, A)
    torch.testing.assert_close(logits.grad, want, rtol=1e-4, atol=1e-5)


@settings(max_examples=20, deadline=None)
@given(T=st.integers(1, 10), B=st.integers(1, 5),
       seed=st.integers(0, 2**31 - 1))
def test_baseline_semi_gradient(T, B, seed):
    """d baseline / d values = (values - vs); vs carries NO gradient
    (the semi-gradient property the learner depends on)."""
    g = torch.Generator().manual_seed(seed)
    vs = torch.randn(T, B, generator=g).requires_grad_(True)
    values = torch.randn(T, B, generator=g).requires_grad_(True)
    losses.baseline_loss(vs, values).backward()
    torch.testing.assert_close(values.grad, values.detach() - vs.detach(),
                               rtol=1e-5, atol=1e-6)
    assert vs.grad is None or torch.all(vs.grad == 0)


@settings(max_examples=20, deadline=None)
@given(A=st.integers(2, 30), seed=st.integers(0, 2**31 - 1))
def test_entropy_bounds(A, seed):
    """Mean entropy (= -entropy_loss / rows) lies in [0, log A]."""
    import math
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(3, 2, A, generator=g) * 5
    mean_ent = -losses.entropy_loss(logits).item() / (3 * 2)
    assert -1e-5 <= mean_ent <= math.log(A) + 1e-5
