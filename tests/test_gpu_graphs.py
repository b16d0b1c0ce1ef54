"""hipGraph-captured learn step == eager learn steps (bit-for-bit device
math on identical staged batches) — marked gpu."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from torched_impala_amd.config import Flags
from torched_impala_amd import learner as L
from torched_impala_amd.models import build_model
from torched_impala_amd.ops.optim import FlatRMSprop


class _StubBatcher:
    """Quacks like Batcher for GraphedStep: staging dict + _to_time_major."""

    def __init__(self, flags, staging):
        self.flags = flags
        self.k = flags.batch_size // flags.envs_per_actor
        self.staging = {0: staging}

    _to_time_major = L.Batcher._to_time_major


def _staging(flags, obs, A, device, seed=0):
    torch.manual_seed(seed)
    T = flags.unroll_length
    k, n = flags.batch_size, 1
    st = dict(
        frame=torch.randint(0, 256, (k, T + 1, n, *obs), dtype=torch.uint8,
                            device=device),
        reward=torch.randn(k, T + 1, n, device=device),
        done=torch.rand(k, T + 1, n, device=device) < 0.05,
        action=torch.randint(0, A, (k, T + 1, n), device=device),
        last_action=torch.randint(0, A, (k, T + 1, n), device=device),
        behavior_logits=torch.randn(k, T + 1, n, A, device=device),
        episode_return=torch.zeros(k, T + 1, n, device=device),
        episode_step=torch.zeros(k, T + 1, n, dtype=torch.int32,
                                 device=device),
    )
    if flags.use_lstm:
        st["initial_core_state"] = torch.zeros(k, 2, n, 256, device=device)
    return st


@pytest.mark.parametrize("use_lstm", [False, True])
@pytest.mark.timeout(600)
def test_graphed_step_matches_eager(use_lstm):
    device = torch.device("cuda", 0)
    obs, A = (4, 84, 84), 6
    flags = Flags(env="synthetic-atari", batch_size=8, unroll_length=10,
                  use_lstm=use_lstm, total_steps=1 << 50)
    n_steps = 2
    # warmup updates are rolled back before capture (ADVICE r1 fix), so
    # the graphed run applies exactly n_steps parameter updates
    total_updates = n_steps

    def make():
        torch.manual_seed(42)
        model = build_model(flags, obs, A).to(device)
        opt, sched = L.build_optimizer(flags, model)
        assert isinstance(opt, FlatRMSprop)
        return model, opt, sched

    # graphed run: first step does WARMUP updates + 1 replay, then replays
    model_g, opt_g, sched_g = make()
    st = _staging(flags, obs, A, device)
    stub = _StubBatcher(flags, st)
    graphed = L.GraphedStep(flags, model_g, opt_g, stub)
    for _ in range(n_steps):
        stats_g = graphed.step(0, sched_g)
    torch.cuda.synchronize()

    # eager run: same number of parameter updates on the same batch
    model_e, opt_e, sched_e = make()
    batch = stub._to_time_major(st)
    for _ in range(total_updates):
        stats_e = L.learn_step(flags, model_e, dict(batch), opt_e, sched_e,
                               None)
    torch.cuda.synchronize()

    torch.testing.assert_close(opt_g.flat_param, opt_e.flat_param,
                               rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(opt_g.flat_sqavg, opt_e.flat_sqavg,
                               rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(stats_g["total_loss"], stats_e["total_loss"],
                               rtol=1e-4, atol=1e-4)


@pytest.mark.timeout(600)
def test_graphed_two_buffers_shared_pool():
    """Both staging buffers get graphs; replays alternate correctly."""
    device = torch.device("cuda", 0)
    obs, A = (4, 84, 84), 6
    flags = Flags(env="synthetic-atari", batch_size=4, unroll_length=6,
                  use_lstm=True, total_steps=1 << 50)
    torch.manual_seed(0)
    model = build_model(flags, obs, A).to(device)
    opt, sched = L.build_optimizer(flags, model)
    st0 = _staging(flags, obs, A, device, seed=1)
    st1 = _staging(flags, obs, A, device, seed=2)
    stub = _StubBatcher(flags, st0)
    stub.staging[1] = st1
    graphed = L.GraphedStep(flags, model, opt, stub)
    losses = []
    for i in (0, 1, 0, 1, 0):
        stats = graphed.step(i, sched)
        losses.append(float(stats["total_loss"]))
    assert all(x == x for x in losses), f"NaN loss in {losses}"
    assert len(graphed.graphs) == 2


@pytest.mark.timeout(600)
def test_graphed_step_world2_stub_equals_world1(monkeypatch):
    """The exact world>1 step sequence (graph replay -> flat all-reduce
    -> eager fused step, learner.py GraphedStep.step) on ONE GPU with a
    stubbed collective simulating a second identical rank (flat_grad
    *= 2 against the optimizer's 1/world scale): must track the world=1
    path step for step.

    Tolerances, not bitwise: the conv-backward dW kernels accumulate
    with float atomics, so even two world=1 runs differ run-to-run
    (measured max |d| 7.5e-9 after 2 steps on hardware); a protocol bug
    (missing scale, missing reduce) diverges orders of magnitude
    faster. The deterministic world-scale algebra is asserted bitwise
    in test_world_scale_optimizer_algebra."""
    monkeypatch.setenv("IMPALA_SHALLOW_MFMA", "0")  # fp32 encoder: lowest noise
    device = torch.device("cuda", 0)
    obs, A = (4, 84, 84), 6
    flags = Flags(env="synthetic-atari", batch_size=8, unroll_length=10,
                  use_lstm=False, total_steps=1 << 50)
    n_steps = 4

    def run(world):
        torch.manual_seed(42)
        model = build_model(flags, obs, A).to(device)
        opt, sched = L.build_optimizer(flags, model, world_size=world)
        st = _staging(flags, obs, A, device)
        stub = _StubBatcher(flags, st)
        graphed = L.GraphedStep(flags, model, opt, stub, world_size=world)
        traj = []
        for _ in range(n_steps):
            graphed.step(0, sched)
            torch.cuda.synchronize()
            traj.append(opt.flat_param.clone())
        graphed.destroy()
        return traj

    traj1 = run(1)

    import torch.distributed as dist

    class _Work:
        def wait(self):
            pass

    calls = []

    def fake_all_reduce(t, op=None, group=None, async_op=False):
        calls.append(tuple(t.shape))
        t.mul_(2.0)  # the second identical rank's contribution (SUM)
        return _Work()

    monkeypatch.setattr(dist, "all_reduce", fake_all_reduce)
    traj2 = run(2)
    assert len(calls) == n_steps, "flat all-reduce must run once per step"
    for i, (a, b) in enumerate(zip(traj1, traj2)):
        torch.testing.assert_close(
            a, b, rtol=5e-3, atol=1e-4,
            msg=lambda m, i=i: f"step {i} diverged between world=1 "
                               f"and stubbed world=2: {m}")


@pytest.mark.timeout(300)
def test_world_scale_optimizer_algebra():
    """Deterministic half of the DP equality claim: for IDENTICAL
    per-rank gradients g, the world=2 sequence (SUM all-reduce -> 2g,
    optimizer scale 1/2) must produce the same parameters as world=1 on
    g. Grads are kept tiny so the global-norm clip coefficient clamps
    to 1.0 on both paths (the norm reduction itself uses atomics) and
    sqrt(sqavg) ~ eps — the regime most sensitive to a scale bug."""
    device = torch.device("cuda", 0)
    torch.manual_seed(7)
    init = torch.randn(4096, device=device)

    def make(world):
        p = torch.nn.Parameter(init.clone())
        opt = FlatRMSprop([p], lr=0.01, max_grad_norm=40.0,
                          world_size=world)
        return opt

    opt1, opt2 = make(1), make(2)
    for it in range(5):
        torch.manual_seed(100 + it)
        g = 1e-3 * torch.randn(4096, device=device)
        opt1.flat_grad.copy_(g)
        opt1.step()
        opt2.flat_grad.copy_(g)
        opt2.flat_grad.mul_(2.0)  # the stubbed SUM all-reduce
        opt2.step()
    torch.cuda.synchronize()
    torch.testing.assert_close(opt1.flat_param, opt2.flat_param,
                               rtol=1e-6, atol=1e-8)
    torch.testing.assert_close(opt1.flat_sqavg, opt2.flat_sqavg,
                               rtol=1e-6, atol=1e-12)
