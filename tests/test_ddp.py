"""Learner DP without GPUs: gloo backend, world_size 2,
on 127.0.0.1. Verifies GradReducer produces the mean gradient and that
ranks stay in lockstep."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from torched_impala_amd.models import MLPPolicy
from torched_impala_amd.parallel import GradReducer


def _worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)  # same init on both ranks
        model = MLPPolicy((4,), 2)
        reducer = GradReducer(model, bucket_bytes=1 << 10)  # force >1 bucket
        torch.manual_seed(100 + rank)  # different data per rank
        x = torch.randn(3, 2, 4)
        inputs = dict(frame=x, reward=torch.zeros(3, 2),
                      done=torch.zeros(3, 2, dtype=torch.bool),
                      last_action=torch.zeros(3, 2, dtype=torch.long))
        reducer.prepare()
        out, _ = model(inputs, ())
        loss = out["policy_logits"].pow(2).sum() + out["baseline"].pow(2).sum()
        loss.backward()
        reducer.finalize()
        grads = torch.cat([p.grad.flatten() for p in model.parameters()])
        q.put((rank, loss.item(), grads.numpy().copy()))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_gradreducer_gloo_world2():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29611
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, loss, grads = q.get(timeout=120)
        results[rank] = (loss, grads)
    for p in procs:
        p.join(timeout=30)
    g0, g1 = results[0][1], results[1][1]
    # after all-reduce(mean) both ranks hold identical gradients
    torch.testing.assert_close(torch.from_numpy(g0), torch.from_numpy(g1), rtol=1e-6, atol=1e-7)


def _worker_ref(rank, world, port, q):
    """Compare reduced grads against a single-process sum/2 reference."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)
        model = MLPPolicy((4,), 2)
        reducer = GradReducer(model)
        x = torch.full((1, 1, 4), float(rank + 1))
        inputs = dict(frame=x, reward=torch.zeros(1, 1),
                      done=torch.zeros(1, 1, dtype=torch.bool),
                      last_action=torch.zeros(1, 1, dtype=torch.long))
        reducer.prepare()
        out, _ = model(inputs, ())
        out["baseline"].sum().backward()
        reducer.finalize()
        if rank == 0:
            q.put(torch.cat([p.grad.flatten() for p in model.parameters()]).numpy().copy())
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_gradreducer_matches_mean_of_local_grads():
    # local reference: mean of grads computed with x=1 and x=2
    torch.manual_seed(0)
    ref_model = MLPPolicy((4,), 2)
    ref_grads = []
    for val in (1.0, 2.0):
        ref_model.zero_grad()
        inputs = dict(frame=torch.full((1, 1, 4), val),
                      reward=torch.zeros(1, 1),
                      done=torch.zeros(1, 1, dtype=torch.bool),
                      last_action=torch.zeros(1, 1, dtype=torch.long))
        out, _ = ref_model(inputs, ())
        out["baseline"].sum().backward()
        ref_grads.append(torch.cat(
            [(p.grad if p.grad is not None else torch.zeros_like(p)).flatten()
             for p in ref_model.parameters()]))
    expected = (ref_grads[0] + ref_grads[1]) / 2

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_ref, args=(r, 2, 29612, q))
             for r in range(2)]
    for p in procs:
        p.start()
    got = q.get(timeout=120)
    for p in procs:
        p.join(timeout=30)
    torch.testing.assert_close(torch.from_numpy(got), expected, rtol=1e-6, atol=1e-7)


def _worker_flat(rank, world, port, q):
    """FlatRMSprop + FlatGradReducer: ranks converge to identical params."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from torched_impala_amd.ops.optim import FlatRMSprop
        from torched_impala_amd.parallel import FlatGradReducer
        torch.manual_seed(0)
        model = MLPPolicy((4,), 2)
        opt = FlatRMSprop(model.parameters(), lr=0.01, max_grad_norm=40.0,
                          world_size=world)
        reducer = FlatGradReducer(model, opt, bucket_bytes=1 << 10)
        for it in range(3):
            torch.manual_seed(1000 * (rank + 1) + it)  # different data
            x = torch.randn(2, 3, 4)
            inputs = dict(frame=x, reward=torch.zeros(2, 3),
                          done=torch.zeros(2, 3, dtype=torch.bool),
                          last_action=torch.zeros(2, 3, dtype=torch.long))
            opt.zero_grad()
            reducer.prepare()
            out, _ = model(inputs, ())
            (out["policy_logits"].pow(2).sum()
             + out["baseline"].pow(2).sum()).backward()
            reducer.finalize()
            opt.step()
        q.put((rank, opt.flat_param.numpy().copy()))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_flat_reducer_gloo_world2():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_flat, args=(r, 2, 29617, q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(2):
        rank, flat = q.get(timeout=120)
        res[rank] = flat
    for p in procs:
        p.join(timeout=30)
    # identical data-parallel updates on both ranks
    torch.testing.assert_close(torch.from_numpy(res[0]),
                               torch.from_numpy(res[1]),
                               rtol=1e-6, atol=1e-7)


def _worker_flat_ref(rank, world, port, q):
    """Flat DP world-2 result == single-process run on the combined data."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from torched_impala_amd.ops.optim import FlatRMSprop
        from torched_impala_amd.parallel import FlatGradReducer
        torch.manual_seed(0)
        model = MLPPolicy((4,), 2)
        opt = FlatRMSprop(model.parameters(), lr=0.01, max_grad_norm=0.0,
                          world_size=world)
        reducer = FlatGradReducer(model, opt)
        x = torch.full((1, 1, 4), float(rank + 1))
        inputs = dict(frame=x, reward=torch.zeros(1, 1),
                      done=torch.zeros(1, 1, dtype=torch.bool),
                      last_action=torch.zeros(1, 1, dtype=torch.long))
        opt.zero_grad()
        reducer.prepare()
        out, _ = model(inputs, ())
        out["baseline"].sum().backward()
        reducer.finalize()
        opt.step()
        if rank == 0:
            q.put(opt.flat_param.numpy().copy())
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_flat_reducer_equals_mean_gradient_update():
    from torched_impala_amd.ops.optim import FlatRMSprop
    # reference: single process, grad = mean of the two ranks' grads
    torch.manual_seed(0)
    ref = MLPPolicy((4,), 2)
    opt_ref = FlatRMSprop(ref.parameters(), lr=0.01, max_grad_norm=0.0)
    grads = []
    for val in (1.0, 2.0):
        opt_ref.zero_grad()
        inputs = dict(frame=torch.full((1, 1, 4), val),
                      reward=torch.zeros(1, 1),
                      done=torch.zeros(1, 1, dtype=torch.bool),
                      last_action=torch.zeros(1, 1, dtype=torch.long))
        out, _ = ref(inputs, ())
        out["baseline"].sum().backward()
        grads.append(opt_ref.flat_grad.clone())
    opt_ref.flat_grad.copy_((grads[0] + grads[1]) / 2)
    opt_ref.step()

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_flat_ref, args=(r, 2, 29618, q))
             for r in range(2)]
    for p in procs:
        p.start()
    got = q.get(timeout=120)
    for p in procs:
        p.join(timeout=30)
    torch.testing.assert_close(torch.from_numpy(got), opt_ref.flat_param,
                               rtol=1e-6, atol=1e-7)


def _resume_worker(rank, world, port, savedir, q):
    """Phase-2 worker: resume a world-2 run from a world-1 checkpoint."""
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from torched_impala_amd import learner
    from torched_impala_amd.config import Flags
    flags = Flags(env="cartpole", num_actors=1, batch_size=4,
                  unroll_length=10, total_steps=10**9, device="cpu",
                  xpid="resume2", savedir=savedir, seed=3 + rank)
    stats = learner.train(flags, max_learner_steps=3)
    q.put((rank, stats["step_frames"], stats["param_sum"]))


@pytest.mark.timeout(300)
def test_multirank_resume_lockstep(tmp_path):
    """every rank must load the checkpoint on resume —
    otherwise rank 0 trains from the checkpoint while the others train
    from fresh init, all-reducing across divergent models and exiting
    the loop at different step counts. Asserts: both ranks resume at the
    checkpoint step and end bitwise-identical."""
    from torched_impala_amd import learner
    from torched_impala_amd.config import Flags
    savedir = str(tmp_path)
    # phase 1: single-process run writes savedir/resume2/model.tar
    flags = Flags(env="cartpole", num_actors=1, batch_size=4,
                  unroll_length=10, total_steps=10**9, device="cpu",
                  xpid="resume2", savedir=savedir, seed=11)
    s1 = learner.train(flags, max_learner_steps=4)
    assert os.path.exists(os.path.join(savedir, "resume2", "model.tar"))
    base_frames = s1["step_frames"]
    assert base_frames == 4 * 4 * 10

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_resume_worker,
                         args=(r, 2, 29621, savedir, q)) for r in range(2)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(2):
        rank, frames, psum = q.get(timeout=240)
        res[rank] = (frames, psum)
    for p in procs:
        p.join(timeout=60)
    # both ranks resumed from the SAME step (base + 3 steps * 40 frames * 2 ranks)
    assert res[0][0] == res[1][0] == base_frames + 3 * 4 * 10 * 2, res
    # and their models stayed in lockstep (bitwise-equal param digests)
    assert res[0][1] == res[1][1], res


def _ramp_worker(rank, world, port, q):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from torched_impala_amd import learner
    from torched_impala_amd.config import Flags
    flags = Flags(env="cartpole", num_actors=1, batch_size=4,
                  unroll_length=10, total_steps=10**9, device="cpu",
                  disable_checkpoint=True, ramp_ready=True, seed=5 + rank)
    stats = learner.train(flags, max_learner_steps=2)
    q.put((rank, stats["learner_steps"]))


@pytest.mark.timeout(300)
def test_ramp_barrier_world2_no_deadlock():
    """bench.py sets ramp_ready on every rank; the ring-full wait + the
    cross-rank barrier must not deadlock under world 2 (the driver's
    SCALE run takes exactly this path)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_ramp_worker, args=(r, 2, 29631, q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(2):
        rank, steps = q.get(timeout=240)
        res[rank] = steps
    for p in procs:
        p.join(timeout=60)
    assert res == {0: 2, 1: 2}, res
