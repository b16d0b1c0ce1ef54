"""IMPALA policy/value networks.

Three model families (architecture constants from
the IMPALA paper, Espeholt et al. 2018):

* ``MLPPolicy``    — CartPole-class: 2x64 hidden, policy + baseline heads.
* ``ShallowNet``   — the "shallow" IMPALA net: conv 16x8x8/s4,
  32x4x4/s2, 32x3x3/s1 -> FC-256 -> heads, optional LSTMCell(256) core.
* ``DeepResNet``   — the 15-layer "deep" IMPALA net: 3 sections of
  [conv3x3 -> maxpool3x3/s2 -> 2 residual blocks] with 16/32/32 channels.

Shared conventions (the frozen API/checkpoint contract):

* forward consumes a dict ``inputs`` of time-major tensors —
  ``frame`` (T, B, *obs), ``reward`` (T, B), ``done`` (T, B) bool,
  ``last_action`` (T, B) — plus a core_state tuple, and returns
  ``(dict(policy_logits=(T,B,A), baseline=(T,B), action=(T,B)), core_state)``.
* the core input is FC-output ++ clipped reward ++ one-hot(last_action)
  (the canonical IMPALA conditioning).
* parameter names are frozen for the checkpoint format:
  conv1/conv2/conv3/fc/core/policy/baseline (shallow),
  fc1/fc2/policy/baseline (MLP), sections.N.* (deep).
"""
from __future__ import annotations

from typing import Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


def _flatten_time(x: torch.Tensor) -> torch.Tensor:
    return x.reshape(-1, *x.shape[2:])


class _CoreMixin(nn.Module):
    """LSTM-or-identity core over the unroll, with done-mask state resets."""

    def _init_core(self, core_input_size: int, use_lstm: bool, hidden: int = 256):
        self.use_lstm = use_lstm
        self.core_input_size = core_input_size
        if use_lstm:
            self.core = nn.LSTMCell(core_input_size, hidden)
            self.core_output_size = hidden
        else:
            self.core_output_size = core_input_size

    def initial_state(self, batch_size: int) -> Tuple[torch.Tensor, ...]:
        if not self.use_lstm:
            return tuple()
        p = next(self.parameters())
        return tuple(
            torch.zeros(batch_size, self.core.hidden_size,
                        device=p.device, dtype=p.dtype)
            for _ in range(2))

    def _run_core(self, core_input: torch.Tensor, done: torch.Tensor,
                  core_state: Tuple[torch.Tensor, ...]):
        """core_input: (T, B, F); done: (T, B) bool -> (T*B, out), state."""
        if not self.use_lstm:
            return _flatten_time(core_input), tuple()
        T, B = core_input.shape[:2]
        if (core_input.is_cuda and T > 1 and self.core.hidden_size == 256
                and not getattr(self, "force_eager_core", False)):
            from torched_impala_amd import ops
            if ops.hip_available():
                from torched_impala_amd.ops.lstm import lstm_core_forward
                return lstm_core_forward(self.core, core_input, done,
                                         core_state)
        notdone = (~done).to(core_input.dtype)
        outputs = []
        h, c = core_state
        for t in range(T):
            nd = notdone[t].unsqueeze(-1)
            h = h * nd
            c = c * nd
            h, c = self.core(core_input[t], (h, c))
            outputs.append(h)
        return torch.cat(outputs, dim=0), (h, c)


class _HeadsMixin(nn.Module):
    def _init_heads(self, in_features: int, num_actions: int):
        self.policy = nn.Linear(in_features, num_actions)
        self.baseline = nn.Linear(in_features, 1)
        self.num_actions = num_actions

    def _heads(self, core_output: torch.Tensor, T: int, B: int, greedy: bool,
               need_action: bool = True):
        policy_logits = self.policy(core_output)
        baseline = self.baseline(core_output)
        out = dict(
            policy_logits=policy_logits.view(T, B, self.num_actions).float(),
            baseline=baseline.view(T, B).float(),
        )
        if need_action:  # the learner skips sampling (keeps RNG out of graphs)
            if greedy:
                action = torch.argmax(policy_logits, dim=-1)
            else:
                action = torch.multinomial(
                    F.softmax(policy_logits.float(), dim=-1),
                    num_samples=1).squeeze(-1)
            out["action"] = action.view(T, B)
        return out


def _core_extras(inputs, num_actions: int, dtype: torch.dtype):
    """clipped reward (T,B,1) ++ one-hot last action (T,B,A)."""
    reward = torch.clamp(inputs["reward"], -1.0, 1.0).unsqueeze(-1).to(dtype)
    last_action = F.one_hot(inputs["last_action"].long(), num_actions).to(dtype)
    return reward, last_action


class MLPPolicy(_CoreMixin, _HeadsMixin):
    """CartPole-class MLP actor-critic (BASELINE.json:7)."""

    def __init__(self, obs_shape=(4,), num_actions=2, use_lstm=False, hidden=64):
        nn.Module.__init__(self)
        self.obs_shape = tuple(obs_shape)
        in_dim = 1
        for d in obs_shape:
            in_dim *= d
        self.fc1 = nn.Linear(in_dim, hidden)
        self.fc2 = nn.Linear(hidden, hidden)
        core_in = hidden + num_actions + 1
        self._init_core(core_in, use_lstm)
        self._init_heads(self.core_output_size, num_actions)

    def forward(self, inputs, core_state=(), greedy: bool = False,
                need_action: bool = True):
        frame = inputs["frame"]
        T, B = frame.shape[:2]
        x = _flatten_time(frame).float()
        x = F.relu(self.fc1(x))
        x = F.relu(self.fc2(x))
        reward, last_action = _core_extras(inputs, self.num_actions, x.dtype)
        core_input = torch.cat(
            [x.view(T, B, -1), reward, last_action], dim=-1)
        core_output, core_state = self._run_core(core_input, inputs["done"], core_state)
        return self._heads(core_output, T, B, greedy, need_action), core_state


class ShallowNet(_CoreMixin, _HeadsMixin):
    """The shallow IMPALA ConvNet (BASELINE.json:8; paper Fig. 3 left).

    conv1 16x8x8/s4 -> conv2 32x4x4/s2 -> conv3 32x3x3/s1 -> FC-256.
    For 84x84 input: 84->20->9->7, flatten 32*7*7=1568.
    """

    def __init__(self, obs_shape=(4, 84, 84), num_actions=6, use_lstm=False):
        nn.Module.__init__(self)
        self.obs_shape = tuple(obs_shape)
        c, h, w = obs_shape
        self.conv1 = nn.Conv2d(c, 16, 8, stride=4)
        self.conv2 = nn.Conv2d(16, 32, 4, stride=2)
        self.conv3 = nn.Conv2d(32, 32, 3, stride=1)

        def osize(s, k, st):
            return (s - k) // st + 1
        h3 = osize(osize(osize(h, 8, 4), 4, 2), 3, 1)
        w3 = osize(osize(osize(w, 8, 4), 4, 2), 3, 1)
        self.flat = 32 * h3 * w3
        self.fc = nn.Linear(self.flat, 256)
        core_in = 256 + num_actions + 1
        self._init_core(core_in, use_lstm)
        self._init_heads(self.core_output_size, num_actions)

    def encode(self, x: torch.Tensor) -> torch.Tensor:
        """(N, C, H, W) uint8/float -> (N, 256). The HIP-kernel hot path."""
        if (x.is_cuda and x.dtype == torch.uint8
                and self.obs_shape == (4, 84, 84)
                and not getattr(self, "force_eager_core", False)):
            from torched_impala_amd import ops
            if ops.hip_available():
                from torched_impala_amd.ops.linear import linear_relu
                from torched_impala_amd.ops.shallow_mfma import \
                    shallow_mfma_enabled
                if shallow_mfma_enabled():
                    # bf16 MFMA encoder + rocBLAS bf16 fc
                    # (IMPALA_SHALLOW_MFMA=1; validation gate in
                    # ops/shallow_mfma.py)
                    from torched_impala_amd.ops.shallow_mfma import \
                        shallow_mfma_features
                    return shallow_mfma_features(self, x)
                from torched_impala_amd.ops.encoder import shallow_encode_conv
                return linear_relu(self.fc, shallow_encode_conv(self, x))
        x = x.float() / 255.0
        x = F.relu(self.conv1(x))
        x = F.relu(self.conv2(x))
        x = F.relu(self.conv3(x))
        return F.relu(self.fc(x.flatten(1)))

    def forward(self, inputs, core_state=(), greedy: bool = False,
                need_action: bool = True):
        frame = inputs["frame"]
        T, B = frame.shape[:2]
        x = self.encode(_flatten_time(frame))
        reward, last_action = _core_extras(inputs, self.num_actions, x.dtype)
        core_input = torch.cat([x.view(T, B, -1), reward, last_action], dim=-1)
        core_output, core_state = self._run_core(core_input, inputs["done"], core_state)
        return self._heads(core_output, T, B, greedy, need_action), core_state


class _ResBlock(nn.Module):
    def __init__(self, ch):
        super().__init__()
        self.conv0 = nn.Conv2d(ch, ch, 3, padding=1)
        self.conv1 = nn.Conv2d(ch, ch, 3, padding=1)

    def forward(self, x):
        y = self.conv0(F.relu(x))
        y = self.conv1(F.relu(y))
        return x + y


class _ResSection(nn.Module):
    def __init__(self, cin, cout):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, 3, padding=1)
        self.res0 = _ResBlock(cout)
        self.res1 = _ResBlock(cout)

    def forward(self, x):
        x = self.conv(x)
        x = F.max_pool2d(x, 3, stride=2, padding=1)
        x = self.res0(x)
        x = self.res1(x)
        return x


class DeepResNet(_CoreMixin, _HeadsMixin):
    """The deep (15-layer) IMPALA ResNet (BASELINE.json:11; paper Fig. 3 right).

    Sections 16/32/32 channels, each: conv3x3 + maxpool/2 + 2 res blocks.
    For 72x96 input: spatial /8 -> 9x12, flatten 32*9*12=3456 -> FC-256.
    """

    def __init__(self, obs_shape=(3, 72, 96), num_actions=15, use_lstm=False):
        nn.Module.__init__(self)
        self.obs_shape = tuple(obs_shape)
        c, h, w = obs_shape
        chans = [16, 32, 32]
        secs, cin = [], c
        for cout in chans:
            secs.append(_ResSection(cin, cout))
            cin = cout
        self.sections = nn.ModuleList(secs)

        def psize(s):
            return (s + 1) // 2
        for _ in chans:
            h, w = psize(h), psize(w)
        self.flat = chans[-1] * h * w
        self.fc = nn.Linear(self.flat, 256)
        core_in = 256 + num_actions + 1
        self._init_core(core_in, use_lstm)
        self._init_heads(self.core_output_size, num_actions)

    def encode(self, x: torch.Tensor) -> torch.Tensor:
        """(N, 3, H, W) uint8/float -> (N, 256).  On GPU the whole conv
        stack runs on the NHWC bf16 MFMA kernels (ops/deepconv.py,
        design docs/deep_conv_design.md); eager fp32 NCHW otherwise."""
        if (x.is_cuda and x.dtype == torch.uint8
                and self.obs_shape[0] == 3
                and not getattr(self, "force_eager_core", False)):
            from torched_impala_amd import ops
            if ops.hip_available():
                from torched_impala_amd.ops.deepconv import deep_encode_train
                return deep_encode_train(self, x)
        x = x.float() / 255.0
        for sec in self.sections:
            x = sec(x)
        x = F.relu(x)
        return F.relu(self.fc(x.flatten(1)))

    def forward(self, inputs, core_state=(), greedy: bool = False,
                need_action: bool = True):
        frame = inputs["frame"]
        T, B = frame.shape[:2]
        x = self.encode(_flatten_time(frame))
        reward, last_action = _core_extras(inputs, self.num_actions, x.dtype)
        core_input = torch.cat([x.view(T, B, -1), reward, last_action], dim=-1)
        core_output, core_state = self._run_core(core_input, inputs["done"], core_state)
        return self._heads(core_output, T, B, greedy, need_action), core_state


def build_model(flags, obs_shape, num_actions):
    kind = flags.model
    if kind == "auto":
        if len(obs_shape) == 1:
            kind = "mlp"
        elif flags.env == "synthetic-dmlab":
            kind = "deep"
        else:
            kind = "shallow"
    if kind == "mlp":
        return MLPPolicy(obs_shape, num_actions, use_lstm=flags.use_lstm)
    if kind == "shallow":
        return ShallowNet(obs_shape, num_actions, use_lstm=flags.use_lstm)
    if kind == "deep":
        return DeepResNet(obs_shape, num_actions, use_lstm=flags.use_lstm)
    raise ValueError(f"unknown model {kind!r}")
