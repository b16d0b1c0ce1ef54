"""CPU actor process (BASELINE.json:5 "CPU actor processes").

Each actor owns ``n = envs_per_actor`` environment instances, a local CPU
copy of the policy (refreshed from the learner's WeightMirror), and
writes whole unroll slots into the shared TrajectoryBuffers:

  claim free slot -> (maybe) pull weights -> roll T steps -> publish full.

Slots have shape (T+1, n, ...). Row semantics (TrajectoryBuffers
docstring): row 0 of each unroll is the carried-over last row of the
previous unroll, and ``initial_core_state`` is the LSTM state *before*
the forward that produced that carried row, so the learner can replay
rows 0..T from it.
"""
from __future__ import annotations

import logging
import os

import numpy as np
import torch

from torched_impala_amd.envs import make_env
from torched_impala_amd.models import build_model

log = logging.getLogger(__name__)

ROW_KEYS = ("frame", "reward", "done", "action", "behavior_logits",
            "last_action", "episode_return", "episode_step")


def resolve_actor_tier(requested: str, obs_shape=(4, 84, 84)) -> str:
    """The tier that will actually run on THIS host for the requested
    --actor-jit flag (native-bf16 silently runs the fp32 native tier on
    hosts without avx512bf16; benchmarks must report the achieved tier,
    not the requested one — VERDICT r1)."""
    if requested != "native-bf16":
        return requested
    if tuple(obs_shape) != (4, 84, 84):
        return "frozen"
    try:
        import torched_impala_amd._impala_actor as ext
        probe = ext.ShallowActorNet(0, True)
        return ("native-bf16" if getattr(probe, "bf16_active", False)
                else "native")
    except Exception:
        return "frozen"


class _FlatActorFwd(torch.nn.Module):
    """Feedforward-policy actor forward as one traceable graph: encode ->
    core-input concat -> policy head -> sample. Skips the baseline head
    (actors never use values) and all module-dispatch glue."""

    def __init__(self, net):
        super().__init__()
        self.net = net

    def forward(self, frame, reward, last_action):
        import torch.nn.functional as F
        x = self.net.encode(frame)
        r = torch.clamp(reward, -1.0, 1.0).unsqueeze(-1)
        oh = F.one_hot(last_action, self.net.num_actions).float()
        ci = torch.cat([x, r, oh], dim=-1)
        logits = self.net.policy(ci)
        probs = F.softmax(logits.float(), dim=-1)
        action = torch.multinomial(probs, num_samples=1).squeeze(-1)
        return logits.float(), action


class _FlatActorLstmFwd(torch.nn.Module):
    """Same, for LSTM-core policies: one traced step carrying (h, c) with
    the done-mask reset inlined."""

    def __init__(self, net):
        super().__init__()
        self.net = net

    def forward(self, frame, reward, done, last_action, h, c):
        import torch.nn.functional as F
        x = self.net.encode(frame)
        r = torch.clamp(reward, -1.0, 1.0).unsqueeze(-1)
        oh = F.one_hot(last_action, self.net.num_actions).float()
        ci = torch.cat([x, r, oh], dim=-1)
        nd = (~done).float().unsqueeze(-1)
        h2, c2 = self.net.core(ci, (h * nd, c * nd))
        logits = self.net.policy(h2)
        probs = F.softmax(logits.float(), dim=-1)
        action = torch.multinomial(probs, num_samples=1).squeeze(-1)
        return logits.float(), action, h2, c2


class NativeActorPolicy:
    """AVX-512/AVX2 C++ actor inference (csrc/actor_net.cpp): zero torch
    dispatch, ~1e-7 parity, cheap weight repack on refresh. Covers the
    shallow (4,84,84) net, feedforward or LSTM-256 core."""

    mode = "native"

    def __init__(self, model, num_actions: int, seed: int,
                 bf16: bool = False):
        import torched_impala_amd._impala_actor as ext
        assert model.obs_shape == (4, 84, 84)
        self.use_lstm = bool(getattr(model, "use_lstm", False))
        if self.use_lstm:
            assert model.core.hidden_size == 256
        self._ext = ext.ShallowActorNet(seed, bf16)
        self.bf16_active = bool(getattr(self._ext, "bf16_active", False))
        if bf16 and not self.bf16_active:
            log.info("native actor: avx512bf16 unavailable on this host — "
                     "running the fp32 AVX-512/AVX2 tier instead")
        self._model = model
        self._A = num_actions
        self.refresh()

    def refresh(self):
        m = self._model
        self._ext.load_weights(
            m.conv1.weight.detach(), m.conv1.bias.detach(),
            m.conv2.weight.detach(), m.conv2.bias.detach(),
            m.conv3.weight.detach(), m.conv3.bias.detach(),
            m.fc.weight.detach(), m.fc.bias.detach(),
            m.policy.weight.detach(), m.policy.bias.detach())
        if self.use_lstm:
            self._ext.load_lstm(
                m.core.weight_ih.detach(), m.core.weight_hh.detach(),
                m.core.bias_ih.detach(), m.core.bias_hh.detach(),
                m.policy.weight.detach(), m.policy.bias.detach())

    def __call__(self, *args):
        if self.use_lstm:
            frame, reward, done, last_action, h, c = args
            n = frame.shape[0]
            logits = torch.empty(n, self._A)
            action = torch.empty(n, dtype=torch.int64)
            h2 = h.detach().clone().contiguous()
            c2 = c.detach().clone().contiguous()
            self._ext.forward_lstm(frame.contiguous(), reward.contiguous(),
                                   done.contiguous(),
                                   last_action.contiguous(), h2, c2,
                                   logits, action)
            return logits, action, h2, c2
        frame, reward, last_action = args
        n = frame.shape[0]
        logits = torch.empty(n, self._A)
        action = torch.empty(n, dtype=torch.int64)
        self._ext.forward(frame.contiguous(), reward.contiguous(),
                          last_action.contiguous(), logits, action)
        return logits, action


class FastActorPolicy:
    """torch.jit fast path for feedforward (non-LSTM) actor policies.

    mode "safe":   trace + freeze with parameters PRESERVED — weight
                   pulls propagate (the params are the same tensors);
                   ~1.15x the eager flat forward.
    mode "frozen": full freeze (weights folded as constants, conv+relu
                   fusion etc.; ~1.6x) — caller must refresh() after
                   weight pulls; each refresh costs a few ms + ~50 ms of
                   profiling warmup, so refreshes are rate-limited by
                   flags.actor_jit_refresh claims.
    """

    def __init__(self, model, example, mode: str = "safe",
                 use_lstm: bool = False):
        self.mode = mode
        self.use_lstm = use_lstm
        wrapper_cls = _FlatActorLstmFwd if use_lstm else _FlatActorFwd
        self._wrapper = wrapper_cls(model).eval()
        self._traced = torch.jit.trace(self._wrapper, example,
                                       check_trace=False)
        self._frozen = None
        self.refresh()

    def refresh(self):
        if self.mode == "frozen":
            self._frozen = torch.jit.freeze(self._traced)
        elif self._frozen is None:
            preserved = ["net." + k for k, _ in
                         self._wrapper.net.named_parameters()]
            self._frozen = torch.jit.freeze(self._traced,
                                            preserved_attrs=preserved)

    def __call__(self, *args):
        return self._frozen(*args)


class Rollout:
    """Rollout state over n env copies driven by one policy forward."""

    def __init__(self, flags, actor_index: int, model):
        n = flags.envs_per_actor
        self.n = n
        self.envs = [make_env(flags, seed=flags.seed + 10000 * actor_index + i)
                     for i in range(n)]
        self.model = model
        obs0 = np.stack([e.reset() for e in self.envs])
        self.frame = torch.from_numpy(obs0)
        self.reward = torch.zeros(n)
        self.done = torch.ones(n, dtype=torch.bool)  # first frame: episode start
        self.last_action = torch.zeros(n, dtype=torch.int64)
        self.core_state = model.initial_state(n)
        # episode stats live in numpy: the hot loop updates them
        # vectorised (torch scalar indexing costs ~10 us/env/step)
        self.episode_return = np.zeros(n, dtype=np.float32)
        self.episode_step = np.zeros(n, dtype=np.int32)
        self.core_before = tuple(t.clone() for t in self.core_state)
        self.carry = None
        self.fast = None
        jit_mode = getattr(flags, "actor_jit", "safe")
        use_lstm = getattr(model, "use_lstm", False)
        if jit_mode != "off" and hasattr(model, "encode"):
            try:
                if (jit_mode in ("native", "native-bf16")
                        and getattr(model, "obs_shape", None) == (4, 84, 84)):
                    self.fast = NativeActorPolicy(
                        model, model.num_actions,
                        seed=flags.seed + 77777 * actor_index,
                        bf16=jit_mode == "native-bf16")
                elif jit_mode in ("native", "native-bf16"):
                    # native covers the shallow feedforward net only;
                    # other policies take the traced-frozen path
                    example = ((self.frame, self.reward, self.done,
                                self.last_action, *self.core_state)
                               if use_lstm else
                               (self.frame, self.reward, self.last_action))
                    self.fast = FastActorPolicy(model, example,
                                                mode="frozen",
                                                use_lstm=use_lstm)
                else:
                    example = ((self.frame, self.reward, self.done,
                                self.last_action, *self.core_state)
                               if use_lstm else
                               (self.frame, self.reward, self.last_action))
                    self.fast = FastActorPolicy(model, example,
                                                mode=jit_mode,
                                                use_lstm=use_lstm)
            except Exception:
                log.exception("actor jit fast path unavailable; using eager")
                self.fast = None
        self._advance()  # produce the first carried row

    @torch.inference_mode()
    def _forward(self):
        inputs = dict(
            frame=self.frame.unsqueeze(0),
            reward=self.reward.unsqueeze(0),
            done=self.done.unsqueeze(0),
            last_action=self.last_action.unsqueeze(0),
        )
        out, new_core = self.model(inputs, self.core_state)
        return out, new_core

    @torch.inference_mode()
    def _fast_forward(self):
        if self.fast.use_lstm:
            logits, action, h, c = self.fast(
                self.frame, self.reward, self.done, self.last_action,
                *self.core_state)
            return {"policy_logits": logits.unsqueeze(0),
                    "action": action.unsqueeze(0)}, (h, c)
        logits, action = self.fast(self.frame, self.reward, self.last_action)
        return {"policy_logits": logits.unsqueeze(0),
                "action": action.unsqueeze(0)}, ()

    def _advance(self, out=None, t: int = 0):
        """One timestep: forward -> record row -> step envs.
        out=None: sets self.carry to the recorded row (torch tensors).
        out=dict of NUMPY views (key -> (T+1, n, ...)): writes row t
        directly — one np slice assignment per key instead of a torch
        copy_ dispatch (the per-claim overhead of the torch path was
        measured at 5.2 ms vs 10.7 ms of compute, 33%)."""
        fwd, new_core = (self._fast_forward() if self.fast is not None
                         else self._forward())
        action = fwd["action"][0]
        if out is not None:
            out["frame"][t] = self.frame.numpy()
            out["reward"][t] = self.reward.numpy()
            out["done"][t] = self.done.numpy()
            out["action"][t] = action.numpy()
            out["behavior_logits"][t] = fwd["policy_logits"][0].numpy()
            out["last_action"][t] = self.last_action.numpy()
            out["episode_return"][t] = self.episode_return
            out["episode_step"][t] = self.episode_step
        else:
            self.carry = dict(
                frame=self.frame, reward=self.reward.clone(),
                done=self.done.clone(), action=action,
                behavior_logits=fwd["policy_logits"][0],
                last_action=self.last_action,  # action INPUT to this forward
                episode_return=torch.from_numpy(self.episode_return.copy()),
                episode_step=torch.from_numpy(self.episode_step.copy()),
            )
        finished = self.done
        if finished.any():  # reset stats after the done row was recorded
            fin = finished.numpy()
            self.episode_return[fin] = 0.0
            self.episode_step[fin] = 0
        self.core_state = new_core
        self.last_action = action
        # step all envs with the chosen actions (auto-reset on done);
        # preallocated numpy buffers keep this loop allocation-free.
        # The obs buffer is DOUBLE-buffered: self.carry["frame"] aliases
        # the buffer the previous step wrote, which must stay intact
        # until fill_slot has copied it out.
        if not hasattr(self, "_np_obs"):
            shape = (self.n, *self.envs[0].obs_shape)
            dt = self.envs[0].obs_dtype
            self._np_obs = [np.empty(shape, dtype=dt), np.empty(shape, dtype=dt)]
            self._obs_flip = 0
        self._obs_flip ^= 1
        obs_buf = self._np_obs[self._obs_flip]
        rew = np.empty(self.n, dtype=np.float32)
        done = np.empty(self.n, dtype=bool)
        # NOTE: a vectorised batch-env (numpy row gathers) was measured
        # SLOWER than this loop (0.79 vs 0.70 ms/step, 16 envs): the
        # synthetic step is trivially cheap and the obs copy dominates
        # either way, so the loop stays.
        act = action.tolist()
        for i, env in enumerate(self.envs):
            o, r, d, _ = env.step(act[i])
            if d:
                o = env.reset()
            obs_buf[i] = o
            rew[i] = r
            done[i] = d
        self.episode_return += rew
        self.episode_step += 1
        self.frame = torch.from_numpy(obs_buf)
        self.reward = torch.from_numpy(rew)
        self.done = torch.from_numpy(done)

    def fill_slot(self, slot, T: int):
        """Write rows 0..T (shape (T+1, n, ...)) of one slot, directly
        into the shm arenas through numpy views (single write pass)."""
        np_out = {k: slot[k].numpy() for k in ROW_KEYS}
        for k in ROW_KEYS:
            src = self.carry[k]
            np_out[k][0] = src.numpy() if isinstance(src, torch.Tensor) else src
        if "initial_core_state" in slot and self.core_before:
            # (2, n, hidden)
            slot["initial_core_state"].copy_(torch.stack(self.core_before))
        for t in range(1, T + 1):
            if t == T:
                self.core_before = tuple(x.clone() for x in self.core_state)
            self._advance(out=np_out, t=t)
        # next unroll's row 0 = this unroll's row T; the slot is about to
        # be handed off (and recycled), so carry owns a copy
        self.carry = {k: torch.from_numpy(np_out[k][T].copy())
                      for k in ROW_KEYS}


def act(flags, actor_index: int, free_queue, full_queue, buffers, mirror,
        obs_shape, obs_dtype, num_actions):
    """Actor process entry point."""
    try:
        # measured on the 256-core MI355X box: 1 thread/actor is optimal —
        # torch intra-op threading on these small per-frame convs costs
        # 4-8x aggregate throughput (48k @4t, 25k @8t vs 200k @1t for 32
        # actors). actor_threads stays as an explicit override only.
        torch.set_num_threads(max(1, flags.actor_threads) if
                              flags.actor_threads > 0 else 1)
        torch.manual_seed(flags.seed + actor_index)
        model = build_model(flags, obs_shape, num_actions)
        model.eval()
        version = mirror.pull_into(model)
        roll = Rollout(flags, actor_index, model)
        T = flags.unroll_length
        refresh_every = max(1, getattr(flags, "actor_jit_refresh", 4))
        claims = 0
        refresh_pending = False  # model updated but weights not re-baked
        while True:
            m = free_queue.pop_wait()
            if m < 0:  # abort/sentinel
                break
            new_version = mirror.pull_into(model, last_seen=version)
            if new_version != version:
                refresh_pending = True
                version = new_version
            if (roll.fast is not None
                    and roll.fast.mode in ("frozen", "native")
                    and refresh_pending
                    and claims % refresh_every == 0):
                roll.fast.refresh()  # re-bake/re-pack the weights
                refresh_pending = False
            claims += 1
            roll.fill_slot(buffers.slot(m), T)
            if os.environ.get("IMPALA_DEBUG_NAN"):
                s = buffers.slot(m)
                if not torch.isfinite(s["behavior_logits"]).all():
                    bad = (~torch.isfinite(s["behavior_logits"]))
                    wfin = all(torch.isfinite(p).all()
                               for p in model.parameters())
                    h, c = (roll.core_state if roll.core_state else
                            (torch.zeros(1), torch.zeros(1)))
                    log.error(
                        "NAN-DEBUG actor=%d claim=%d badrows=%d wfinite=%s "
                        "hfin=%s cfin=%s hmax=%.3e cmax=%.3e ver=%d",
                        actor_index, claims, int(bad.any(-1).sum()), wfin,
                        torch.isfinite(h).all().item(),
                        torch.isfinite(c).all().item(),
                        h[torch.isfinite(h)].abs().max().item(),
                        c[torch.isfinite(c)].abs().max().item(), version)
                    dump = f"/tmp/nan_dump_a{actor_index}.pt"
                    if wfin and not os.path.exists(dump):
                        torch.save({"sd": model.state_dict(),
                                    "slot": {k: v.clone()
                                             for k, v in s.items()},
                                    "h": h.clone(), "c": c.clone()}, dump)
            if not full_queue.push(m):
                log.error("actor %d: full queue overflow", actor_index)
                break
    except KeyboardInterrupt:
        pass
    except Exception:
        log.exception("actor %d crashed", actor_index)
        raise
