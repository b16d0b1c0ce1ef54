"""The learner: batch assembly, V-trace learn step, and train()/test()
orchestration.

MI355X specifics:
* batch drain: full slots are copied shm -> HBM with async copies on a
  dedicated side HIP stream (arenas hipHostRegister'ed), overlapping the
  previous step's forward/backward; double-buffered device staging.
* DP: GradReducer (parallel/ddp.py) all-reduces grad buckets over
  RCCL/xGMI as backward produces them.
* losses: ops.fused_vtrace_loss — gfx950 HIP kernels on GPU.
"""
from __future__ import annotations

import logging
import os
import time
from typing import Dict, List, Optional, Tuple

import torch
import torch.multiprocessing as mp

from torched_impala_amd import checkpoint as ckpt_mod
from torched_impala_amd import ops
from torched_impala_amd.actor import act
from torched_impala_amd.config import Flags
from torched_impala_amd.envs import env_spec
from torched_impala_amd.metrics import MetricsLogger, StageTimers
from torched_impala_amd.models import build_model
from torched_impala_amd.ops.optim import FlatRMSprop
from torched_impala_amd.parallel import (FlatGradReducer, GradReducer,
                                         init_distributed)
from torched_impala_amd.transport import SlotQueue, TrajectoryBuffers, WeightMirror

log = logging.getLogger(__name__)


def _np_to_torch_dtype(np_dtype):
    import numpy as np
    return {np.dtype(np.uint8): torch.uint8,
            np.dtype(np.float32): torch.float32}[np.dtype(np_dtype)]


def create_buffers(flags: Flags, obs_shape, obs_dtype, num_actions
                   ) -> TrajectoryBuffers:
    n = flags.envs_per_actor
    assert flags.batch_size % n == 0, "batch_size must divide envs_per_actor"
    slots_per_batch = flags.batch_size // n
    num_slots = flags.num_buffers or max(2 * slots_per_batch,
                                         slots_per_batch + flags.num_actors)
    return TrajectoryBuffers(
        unroll_length=flags.unroll_length,
        obs_shape=obs_shape,
        obs_dtype=_np_to_torch_dtype(obs_dtype),
        num_actions=num_actions,
        num_slots=num_slots,
        lstm_hidden=256 if flags.use_lstm else 0,
        envs_per_slot=n)


class Batcher:
    """Assembles time-major device batches from full ring slots.

    GPU: double-buffered device staging filled by async H2D copies on a
    side stream; slots are freed once their copy event completes.
    CPU: plain stack.
    """

    def __init__(self, flags: Flags, buffers: TrajectoryBuffers,
                 free_queue: SlotQueue, full_queue: SlotQueue, device):
        self.flags = flags
        self.buffers = buffers
        self.free_queue = free_queue
        self.full_queue = full_queue
        self.device = device
        self.k = flags.batch_size // flags.envs_per_actor
        self.is_cuda = device.type == "cuda"
        self._pending: List[Tuple[torch.cuda.Event, List[int]]] = []
        if self.is_cuda:
            buffers.pin()
            self.copy_stream = torch.cuda.Stream()
            self.staging = []
            self.consumed_ev = []
            for _ in range(2):
                st = {key: torch.zeros((self.k, *a.shape[1:]), dtype=a.dtype,
                                       device=device)
                      for key, a in buffers.arenas.items()}
                self.staging.append(st)
                ev = torch.cuda.Event()
                ev.record()  # initially free
                self.consumed_ev.append(ev)
            self.idx = 0

    def _drain_pending(self, block: bool = False):
        still = []
        for ev, slots in self._pending:
            if block:
                ev.synchronize()
            if ev.query():
                for m in slots:
                    self.free_queue.push(m)
            else:
                still.append((ev, slots))
        self._pending = still

    def _pop_slots(self) -> List[int]:
        slots = []
        while len(slots) < self.k:
            # generous: a cold 8-rank start spawns 256 actor processes
            # that each import torch before the first slot lands
            m = self.full_queue.pop_wait(timeout_ms=300000.0)
            if m == -2:
                raise RuntimeError("transport aborted")
            if m < 0:
                raise TimeoutError(
                    "learner starved: no full rollout slot within 300 s "
                    f"(got {len(slots)}/{self.k}) — actors dead or too slow")
            slots.append(m)
        return slots

    def next_batch(self, materialize: bool = True) -> Dict[str, torch.Tensor]:
        """Returns time-major batch: frame (T+1, B, *obs), ...,
        initial_core_state (2, B, H) if present. With materialize=False
        (graph mode) only drains into the staging buffer and returns its
        index — the captured graph owns the time-major views."""
        if not self.is_cuda:
            return self._next_batch_cpu()
        self._drain_pending()
        i = self.idx
        self.idx = (self.idx + 1) % 2
        slots = self._pop_slots()
        st = self.staging[i]
        if os.environ.get("IMPALA_NO_COPY_OVERLAP", "0") == "1":
            # diagnostic: H2D on the compute stream (no side-stream
            # overlap) — race-isolation A/B for the r12-r19 forensics
            for key, arena in self.buffers.arenas.items():
                dst = st[key]
                for j, m in enumerate(slots):
                    dst[j].copy_(arena[m], non_blocking=True)
            copy_done = torch.cuda.Event()
            copy_done.record()
        else:
            with torch.cuda.stream(self.copy_stream):
                self.copy_stream.wait_event(self.consumed_ev[i])
                for key, arena in self.buffers.arenas.items():
                    dst = st[key]
                    for j, m in enumerate(slots):
                        dst[j].copy_(arena[m], non_blocking=True)
                copy_done = torch.cuda.Event()
                copy_done.record(self.copy_stream)
            torch.cuda.current_stream().wait_event(copy_done)
        self._pending.append((copy_done, slots))
        batch = self._to_time_major(st) if materialize else {}
        batch["_staging_index"] = i
        return batch

    def mark_consumed(self, batch: Dict[str, torch.Tensor]):
        """Record that the compute stream is done reading the staging
        buffer backing `batch` (call after backward)."""
        if self.is_cuda:
            i = batch.pop("_staging_index")
            self.consumed_ev[i].record()

    def _to_time_major(self, st: Dict[str, torch.Tensor]):
        out = {}
        k, n = self.k, self.flags.envs_per_actor
        use_kernel = False
        for v in st.values():
            use_kernel = v.is_cuda and ops.hip_available()
            break
        ext = ops.ext_hip() if use_kernel else None
        for key, v in st.items():
            if use_kernel:
                # one block-permutation kernel per tensor instead of an
                # ATen permute+reshape copy (SURVEY §2c rollout_unpack;
                # ~80 -> ~25 us/step inside the captured graph)
                out[key] = ext.unpack_time_major(v.contiguous(), 1)
                continue
            if key == "initial_core_state":
                # (k, 2, n, H) -> (2, k*n, H)
                out[key] = v.permute(1, 0, 2, 3).reshape(2, k * n, -1)
            else:
                # (k, T+1, n, ...) -> (T+1, k*n, ...)
                out[key] = v.transpose(0, 1).reshape(
                    v.shape[1], k * n, *v.shape[3:])
        return out

    def _next_batch_cpu(self):
        slots = self._pop_slots()
        st = {key: torch.stack([arena[m] for m in slots])
              for key, arena in self.buffers.arenas.items()}
        for m in slots:
            self.free_queue.push(m)
        return self._to_time_major(st)

    def shutdown(self):
        self._drain_pending(block=True)


def _fwd_bwd(flags: Flags, model, batch: Dict[str, torch.Tensor],
             optimizer, reducer=None, timers: Optional[StageTimers] = None):
    """Forward + V-trace/loss + backward: pure device work, every op
    hipGraph-capturable (GraphedStep captures exactly this)."""
    t = timers or StageTimers()
    t.start("forward")
    core_state = ()
    if "initial_core_state" in batch:
        ics = batch["initial_core_state"]
        core_state = (ics[0], ics[1])
    inputs = dict(frame=batch["frame"], reward=batch["reward"],
                  done=batch["done"], last_action=batch["last_action"])
    if flags.compute_dtype == "bf16" and batch["frame"].is_cuda:
        # bf16 matmul/conv compute (MFMA bf16 rate); heads re-cast to
        # fp32, so V-trace math stays fp32. Mainly for the deep/big
        # config — the headline benches run fp32.
        with torch.autocast("cuda", dtype=torch.bfloat16):
            outputs, _ = model(inputs, core_state, need_action=False)
    else:
        outputs, _ = model(inputs, core_state, need_action=False)

    logits_all = outputs["policy_logits"]      # (T+1, B, A)
    baseline_all = outputs["baseline"]         # (T+1, B)
    bootstrap_value = baseline_all[-1]
    target_logits = logits_all[:-1]
    values = baseline_all[:-1]
    actions = batch["action"][:-1]
    behavior_logits = batch["behavior_logits"][:-1]
    rewards = batch["reward"][1:]
    if flags.reward_clipping == "abs_one":
        rewards = torch.clamp(rewards, -1.0, 1.0)
    discounts = (~batch["done"][1:]).float() * flags.discounting

    loss, stats, vs, adv = ops.fused_vtrace_loss(
        target_logits, values, bootstrap_value, behavior_logits,
        actions, rewards, discounts,
        clip_rho=flags.clip_rho_threshold, clip_c=flags.clip_c_threshold,
        lambda_=flags.lambda_,
        clip_pg_rho=getattr(flags, "clip_pg_rho_threshold", None),
        baseline_cost=flags.baseline_cost, entropy_cost=flags.entropy_cost,
        force_eager=not flags.use_hip_kernels)

    t.start("backward")
    optimizer.zero_grad(set_to_none=False)
    if reducer:
        reducer.prepare()
    loss.backward()
    t.stop()
    return loss, stats


def learn_step(flags: Flags, model, batch: Dict[str, torch.Tensor],
               optimizer, scheduler, reducer: Optional[GradReducer],
               timers: Optional[StageTimers] = None) -> Dict[str, float]:
    """One IMPALA update on one batch."""
    t = timers or StageTimers()
    loss, stats = _fwd_bwd(flags, model, batch, optimizer, reducer, t)
    t.start("allreduce")
    if reducer:
        reducer.finalize()
    t.start("optimizer")
    if isinstance(optimizer, FlatRMSprop):
        optimizer.step()  # fused: clip-by-global-norm folded in
    else:
        torch.nn.utils.clip_grad_norm_(model.parameters(),
                                       flags.grad_norm_clipping)
        optimizer.step()
    scheduler.step()
    t.stop()
    return dict(total_loss=loss.detach(), **stats)


class GraphedStep:
    """hipGraph capture of the entire learn step (time-major staging views
    -> forward -> fused V-trace loss -> backward -> fused clip+RMSProp),
    one graph per staging buffer, shared memory pool.

    MI355X-native rationale: the headline step is hundreds of small
    kernels over (T+1)*B=672-row tensors — launch overhead dominates;
    one hipGraph replay removes it ("HIP streams and graphs instead of a
    tracing compiler"). lr reaches the captured optimizer through a
    device scalar (FlatRMSprop.sync_lr runs outside the graph).

    Data parallelism (world > 1): the RCCL collective stays OUTSIDE the
    graph — the captured region is forward+loss+backward only (grads
    land in the flat arena), then ONE flat all-reduce of flat_grad and
    the eager fused optimizer step run per replay. This trades the
    backward-overlapped bucketing of FlatGradReducer for graph-level
    launch elimination (the grads are a few MB — latency-dominated
    either way on xGMI).

    Preconditions (asserted by the caller): CUDA, FlatRMSprop, HIP
    kernels on.
    """

    WARMUP_ITERS = 2

    def __init__(self, flags, model, optimizer, batcher, world_size: int = 1):
        self.flags = flags
        self.model = model
        self.optimizer = optimizer
        self.batcher = batcher
        self.world = world_size
        self.graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self.outputs: Dict[int, tuple] = {}
        self.pool = None

    def _fn(self, i: int):
        batch = self.batcher._to_time_major(self.batcher.staging[i])
        loss, stats = _fwd_bwd(self.flags, self.model, batch, self.optimizer)
        if self.world == 1:
            self.optimizer.step_device()  # world>1: runs eagerly after
        return (loss.detach(), stats["pg_loss"], stats["baseline_loss"],
                stats["entropy_loss"])

    def _capture(self, i: int):
        # warmup runs real optimizer updates (world==1 captures
        # step_device inside _fn) — snapshot the flat state so capture
        # doesn't bake in up to 2*WARMUP_ITERS uncounted weight updates
        # from the same batch
        snap = (self.optimizer.flat_param.clone(),
                self.optimizer.flat_sqavg.clone())
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(self.WARMUP_ITERS):
                self._fn(i)
        torch.cuda.current_stream().wait_stream(s)
        with torch.no_grad():
            self.optimizer.flat_param.copy_(snap[0])
            self.optimizer.flat_sqavg.copy_(snap[1])
        del snap
        g = torch.cuda.CUDAGraph()
        ctx = (torch.cuda.graph(g, pool=self.pool) if self.pool is not None
               else torch.cuda.graph(g))
        with ctx:
            self.outputs[i] = self._fn(i)
        if self.pool is None:
            self.pool = g.pool()
        self.graphs[i] = g
        log.info("captured learn-step graph for staging buffer %d", i)

    def destroy(self):
        """Deterministic teardown: the hipGraphExec objects and their
        shared mempool must not be destroyed by a lazy GC while replays
        or side-stream copies are still in flight (suspected source of
        the round-1 flaky async hipErrorInvalidValue)."""
        if self.graphs:
            torch.cuda.synchronize()
        self.graphs.clear()
        self.outputs.clear()
        self.pool = None

    def ensure_captured(self):
        """Capture graphs for ALL staging buffers up-front, BEFORE any
        replay: a capture that shares a mempool with an already-replayed
        graph may recycle blocks that graph's replays still read/write
        (the torch pool-sharing contract is capture-everything-first).
        The lazy per-buffer capture interleaved replays between captures
        and corrupted pool memory data-dependently — NaN losses at
        ~1-2k steps in the r12/r13 soaks, MFMA+graphs only (more live
        pool tensors), and plausibly round-1's flaky async
        hipErrorInvalidValue. IMPALA_GRAPH_LAZY_CAPTURE=1 restores the
        old behavior for A/B."""
        st = self.batcher.staging
        idxs = sorted(st.keys() if isinstance(st, dict) else
                      range(len(st)))
        for j in idxs:
            if j not in self.graphs:
                self._capture(j)

    def step(self, i: int, scheduler) -> Dict[str, torch.Tensor]:
        self.optimizer.sync_lr()
        if i not in self.graphs:
            if os.environ.get("IMPALA_GRAPH_LAZY_CAPTURE", "0") == "1":
                self._capture(i)
            else:
                self.ensure_captured()
        self.graphs[i].replay()
        if self.world > 1:
            # SUM; the 1/world mean is the optimizer's device-side scale
            torch.distributed.all_reduce(self.optimizer.flat_grad,
                                         op=torch.distributed.ReduceOp.SUM)
            self.optimizer.step_device()
        scheduler.step()
        loss, pg, bl, ent = self.outputs[i]
        return dict(total_loss=loss, pg_loss=pg, baseline_loss=bl,
                    entropy_loss=ent)


def build_optimizer(flags: Flags, model, world_size: int = 1):
    if flags.momentum == 0.0:
        optimizer = FlatRMSprop(
            model.parameters(), lr=flags.learning_rate, alpha=flags.alpha,
            eps=flags.epsilon, max_grad_norm=flags.grad_norm_clipping,
            world_size=world_size)
    else:  # momentum needs the torch path (flat kernel is momentum-free)
        optimizer = torch.optim.RMSprop(
            model.parameters(), lr=flags.learning_rate, alpha=flags.alpha,
            momentum=flags.momentum, eps=flags.epsilon)
    frames_per_step = flags.unroll_length * flags.batch_size * world_size

    def lr_lambda(step):
        return max(0.0, 1.0 - step * frames_per_step / flags.total_steps)

    scheduler = torch.optim.lr_scheduler.LambdaLR(optimizer, lr_lambda)
    return optimizer, scheduler


def start_actors(flags: Flags, ctx, buffers, free_queue, full_queue, mirror,
                 obs_shape, obs_dtype, num_actions) -> List[mp.Process]:
    procs = []
    for i in range(flags.num_actors):
        p = ctx.Process(
            target=act,
            args=(flags, i, free_queue, full_queue, buffers, mirror,
                  obs_shape, obs_dtype, num_actions),
            daemon=True, name=f"actor-{i}")
        p.start()
        procs.append(p)
    return procs


def train(flags: Flags, max_learner_steps: Optional[int] = None,
          on_step=None, debug_ctx=None) -> Dict[str, float]:
    """Full actor-learner training. Returns final stats."""
    device = torch.device(flags.run_device)
    rank, world = init_distributed(device.type)
    if device.type == "cuda":
        device = torch.device("cuda", torch.cuda.current_device())
        # don't let the learner's CPU-side OMP pool fight the actors
        torch.set_num_threads(min(8, os.cpu_count() or 8))
    torch.manual_seed(flags.seed + rank)

    obs_shape, obs_dtype, num_actions = env_spec(flags)
    model = build_model(flags, obs_shape, num_actions).to(device)
    model.force_eager_core = not flags.use_hip_kernels
    if world > 1:
        for p in model.parameters():
            torch.distributed.broadcast(p.data, src=0)

    optimizer, scheduler = build_optimizer(flags, model, world)

    cpu_model = build_model(flags, obs_shape, num_actions)
    cpu_model.load_state_dict({k: v.cpu() for k, v in model.state_dict().items()})
    flat_layout = None
    if isinstance(optimizer, FlatRMSprop):
        name_of = {p: n for n, p in model.named_parameters()}
        flat_layout = {name_of[p]: optimizer.slices[p]
                       for p in optimizer._order if p in name_of}
    mirror = WeightMirror(cpu_model, flat_layout=flat_layout)
    use_flat_publish = (mirror.layout is not None
                        and isinstance(optimizer, FlatRMSprop)
                        and device.type == "cuda")
    if use_flat_publish:
        mirror.pin()  # page-lock the flat shm arenas for async D2H

    buffers = create_buffers(flags, obs_shape, obs_dtype, num_actions)
    free_queue = SlotQueue(buffers.num_slots)
    full_queue = SlotQueue(buffers.num_slots)
    for m in range(buffers.num_slots):
        free_queue.push(m)

    ctx = mp.get_context("spawn")
    actor_args = (flags, ctx, buffers, free_queue, full_queue, mirror,
                  obs_shape, obs_dtype, num_actions)
    procs = start_actors(*actor_args)

    batcher = Batcher(flags, buffers, free_queue, full_queue, device)
    graphed = None
    reducer = None
    use_graphs = (flags.graph_capture and device.type == "cuda"
                  and flags.use_hip_kernels
                  and isinstance(optimizer, FlatRMSprop))
    if use_graphs:
        # world>1: the collective runs OUTSIDE the graph (GraphedStep) —
        # no hook-based reducer (its all_reduce calls must not be
        # captured inside backward)
        graphed = GraphedStep(flags, model, optimizer, batcher,
                              world_size=world)
    elif world > 1:
        reducer = (FlatGradReducer(model, optimizer)
                   if isinstance(optimizer, FlatRMSprop)
                   else GradReducer(model))
    metrics = MetricsLogger(flags.metrics_jsonl)
    timers = StageTimers()
    if debug_ctx is not None:
        # diagnostic harnesses (scripts/nan_probe.py --hunt) get direct
        # handles; never used in production paths
        debug_ctx.batcher = batcher
        debug_ctx.optimizer = optimizer
        debug_ctx.model = model

    run_dir = None
    ckpt_path = None
    resume_frames = 0
    if flags.xpid:
        # every rank computes the path and loads the checkpoint: ranks
        # that resumed from different states (or none) would all-reduce
        # gradients across divergent models and exit the step loop at
        # different iteration counts, hanging the others in collectives.
        # Single node => shared filesystem, so all ranks read the same
        # (atomically-replaced) file; rank 0 alone writes it.
        run_dir = os.path.join(os.path.expanduser(flags.savedir), flags.xpid)
        if rank == 0:
            os.makedirs(run_dir, exist_ok=True)
        ckpt_path = os.path.join(run_dir, "model.tar")
        if world > 1:
            torch.distributed.barrier()  # rank0 mkdir before others look
        if os.path.exists(ckpt_path) and not flags.disable_checkpoint:
            ckpt = ckpt_mod.load(ckpt_path, model, optimizer, scheduler,
                                 map_location=device)
            resume_frames = int(ckpt.get("step", 0))
            log.info("rank %d resumed from %s at step=%d",
                     rank, ckpt_path, resume_frames)
            mirror.publish_state_dict(model.state_dict())
        if world > 1:
            # belt and braces: bitwise-identical start even if a rank
            # raced the file; also agree on the resume step.
            if isinstance(optimizer, FlatRMSprop):
                torch.distributed.broadcast(optimizer.flat_param, src=0)
                torch.distributed.broadcast(optimizer.flat_sqavg, src=0)
            else:
                for p in model.parameters():
                    torch.distributed.broadcast(p.data, src=0)
            rf = torch.tensor([resume_frames], dtype=torch.long,
                              device=device)
            torch.distributed.broadcast(rf, src=0)
            resume_frames = int(rf.item())
            if resume_frames:
                mirror.publish_state_dict(model.state_dict())

    if getattr(flags, "ramp_ready", False):
        # wait until the ring is full: every actor has finished at least
        # one claim (imports, weight packing, jit all warmed) before the
        # first learner step — benches then measure steady state even
        # over short windows (73 ms timed region on a 7 s
        # process was ramp-dominated)
        deadline = time.time() + 600.0
        target = buffers.num_slots
        while (full_queue.size_approx() < target
               and time.time() < deadline):
            for i, pr in enumerate(procs):
                if not pr.is_alive():
                    raise RuntimeError(
                        f"actor {i} died during ramp (exit {pr.exitcode})")
            time.sleep(0.02)
        log.info("ramp ready: %d/%d slots full", full_queue.size_approx(),
                 target)
        if world > 1:
            torch.distributed.barrier()

    # debug: IMPALA_DUMP_NONFINITE=<path> saves the batch + optimizer
    # state the first time total_loss goes non-finite (costs one host
    # sync per step; leave unset in production)
    _dump_nonfinite = os.environ.get("IMPALA_DUMP_NONFINITE")
    _dumped = False

    frames_per_batch = flags.unroll_length * flags.batch_size
    step_frames = resume_frames
    learner_steps = 0
    last_ckpt = time.time()
    t_start = time.time()
    last_log = t_start
    last_log_frames = 0
    stats: Dict[str, float] = {}
    try:
        while step_frames < flags.total_steps:
            if max_learner_steps is not None and learner_steps >= max_learner_steps:
                break
            timers.start("rollout_wait")
            batch = batcher.next_batch(materialize=graphed is None)
            si = batch.get("_staging_index")
            if graphed is not None and world > 1 and learner_steps == 0:
                # multi-rank: agree on graph capture BEFORE any training
                # collective — a rank that silently fell back to the
                # bucketed eager path would issue a different collective
                # pattern than the flat-all-reduce graph path and hang
                # the job. One MIN-reduce of the capture outcome keeps
                # every rank on the same path.
                try:
                    graphed.ensure_captured()
                    ok = 1.0
                except Exception:
                    log.exception("graph capture failed on this rank")
                    ok = 0.0
                flag = torch.tensor([ok], device=device)
                torch.distributed.all_reduce(
                    flag, op=torch.distributed.ReduceOp.MIN)
                if float(flag.item()) < 1.0:
                    log.warning("graph capture unavailable on >=1 rank; "
                                "all ranks using eager learn steps")
                    graphed = None
                    reducer = FlatGradReducer(model, optimizer)
            if graphed is not None:
                timers.start("graph_step")
                try:
                    stats = graphed.step(si, scheduler)
                except Exception:
                    log.exception("graph step failed; falling back to "
                                  "eager learn steps")
                    graphed = None
                    if world > 1 and reducer is None:
                        reducer = FlatGradReducer(model, optimizer)
                    batch.update(batcher._to_time_major(batcher.staging[si]))
                    stats = learn_step(flags, model, batch, optimizer,
                                       scheduler, reducer, timers)
                timers.stop()
            else:
                stats = learn_step(flags, model, batch, optimizer, scheduler,
                                   reducer, timers)
            batcher.mark_consumed(batch)
            learner_steps += 1
            step_frames += frames_per_batch * world

            if _dump_nonfinite and not _dumped:
                tl = float(stats["total_loss"])
                if tl != tl or tl == float("inf") or tl == float("-inf"):
                    _dumped = True
                    blob = dict(step=learner_steps,
                                flat_param=optimizer.flat_param.cpu()
                                if isinstance(optimizer, FlatRMSprop)
                                else None,
                                flat_grad=optimizer.flat_grad.cpu()
                                if isinstance(optimizer, FlatRMSprop)
                                else None,
                                flat_sqavg=optimizer.flat_sqavg.cpu()
                                if isinstance(optimizer, FlatRMSprop)
                                else None,
                                stats={k: float(v)
                                       for k, v in stats.items()})
                    if si is not None and batcher.is_cuda:
                        blob["staging"] = {
                            k: v.cpu() for k, v in
                            batcher.staging[si].items()}
                    torch.save(blob, _dump_nonfinite)
                    log.error("non-finite loss at learner step %d — "
                              "dumped state to %s", learner_steps,
                              _dump_nonfinite)

            if learner_steps % flags.weight_sync_interval == 0:
                timers.start("weight_publish")
                if use_flat_publish and os.environ.get(
                        "IMPALA_NO_ASYNC_PUBLISH", "0") != "1":
                    # one D2D snapshot + one async D2H; seqlock flips on
                    # completion (poll) — host cost is an event query
                    mirror.publish_flat_async(optimizer.flat_param)
                else:
                    mirror.publish_state_dict(model.state_dict())
                timers.stop()
            elif use_flat_publish:
                mirror.poll_publish()

            # episode metrics (cheap on CPU; sync-inducing on GPU -> sparse)
            if device.type != "cuda" or learner_steps % 20 == 0:
                if graphed is not None and si is not None:
                    st = batcher.staging[si]
                    done = st["done"][:, 1:]
                    rets = st["episode_return"][:, 1:][done]
                else:
                    done = batch["done"][1:]
                    rets = batch["episode_return"][1:][done]
                if rets.numel():
                    metrics.record_episodes(rets.cpu().tolist())

            # respawn dead actors (SURVEY.md §5c failure detection)
            for i, p in enumerate(procs):
                if not p.is_alive():
                    log.warning("actor %d died (exit %s); respawning",
                                i, p.exitcode)
                    procs[i] = ctx.Process(
                        target=act,
                        args=(flags, i, free_queue, full_queue, buffers,
                              mirror, obs_shape, obs_dtype, num_actions),
                        daemon=True, name=f"actor-{i}")
                    procs[i].start()

            now = time.time()
            if rank == 0 and now - last_log > 5.0:
                fps = (step_frames - last_log_frames) / (now - last_log)
                mr = metrics.mean_return
                log.info("step=%d fps=%.0f loss=%.3f return=%s",
                         step_frames, fps,
                         float(stats["total_loss"]),
                         f"{mr:.2f}" if mr is not None else "n/a")
                metrics.write(step=step_frames, fps=fps,
                              mean_return=mr,
                              timers=timers.drain(),
                              peak_hbm_gib=(round(
                                  torch.cuda.max_memory_allocated() / 2**30, 2)
                                  if device.type == "cuda" else 0.0),
                              queue_full_depth=full_queue.size_approx(),
                              **{k: float(v) for k, v in stats.items()})
                last_log, last_log_frames = now, step_frames

            if (ckpt_path and rank == 0 and not flags.disable_checkpoint
                    and now - last_ckpt > flags.checkpoint_interval_s):
                ckpt_mod.save(ckpt_path, model, optimizer, scheduler,
                              step_frames, flags)
                last_ckpt = now
            if on_step is not None:
                on_step(learner_steps, step_frames, stats)
    finally:
        free_queue.abort()
        full_queue.abort()
        batcher.shutdown()
        if graphed is not None:
            graphed.destroy()
        for p in procs:
            p.join(timeout=5.0)
        for p in procs:
            if p.is_alive():
                p.terminate()
        metrics.close()
    if ckpt_path and rank == 0 and not flags.disable_checkpoint:
        ckpt_mod.save(ckpt_path, model, optimizer, scheduler, step_frames, flags)
    out = {k: float(v) for k, v in stats.items()}
    # deterministic param digest: DP ranks in lockstep agree bitwise
    out["param_sum"] = float(sum(p.data.double().sum().item()
                                 for p in model.parameters()))
    # did the hipGraph step survive the whole run? (a silent eager
    # fallback must never be reported as the graph-path number)
    out["graph_path_active"] = bool(graphed is not None) if use_graphs \
        else False
    if isinstance(optimizer, FlatRMSprop):
        out["grad_skip_count"] = float(optimizer.skip_t.item())
    out.update(step_frames=step_frames, learner_steps=learner_steps,
               mean_return=metrics.mean_return or float("nan"),
               fps=step_frames / max(time.time() - t_start, 1e-9),
               timers={k: round(v, 4) for k, v in timers.drain().items()})
    return out


def test(flags: Flags, num_episodes: int = 10, ckpt_path: Optional[str] = None
         ) -> Dict[str, float]:
    """Eval entry: load checkpoint, run greedy episodes."""
    from torched_impala_amd.envs import make_env
    obs_shape, obs_dtype, num_actions = env_spec(flags)
    model = build_model(flags, obs_shape, num_actions)
    model.eval()
    if ckpt_path is None and flags.xpid:
        ckpt_path = os.path.join(
            os.path.expanduser(flags.savedir), flags.xpid, "model.tar")
    if ckpt_path and os.path.exists(ckpt_path):
        ckpt_mod.load(ckpt_path, model)
    env = make_env(flags, seed=flags.seed)
    returns = []
    with torch.no_grad():
        for ep in range(num_episodes):
            obs = env.reset()
            core_state = model.initial_state(1)
            reward, done, last_action, ep_ret = 0.0, True, 0, 0.0
            while True:
                inputs = dict(
                    frame=torch.from_numpy(obs)[None, None],
                    reward=torch.tensor([[reward]], dtype=torch.float32),
                    done=torch.tensor([[done]]),
                    last_action=torch.tensor([[last_action]]))
                out, core_state = model(inputs, core_state, greedy=True)
                a = int(out["action"][0, 0])
                obs, reward, done, _ = env.step(a)
                ep_ret += reward
                last_action = a
                if done:
                    break
            returns.append(ep_ret)
    log.info("test: %d episodes, mean return %.2f", num_episodes,
             sum(returns) / len(returns))
    return dict(mean_return=sum(returns) / len(returns),
                returns=returns)
