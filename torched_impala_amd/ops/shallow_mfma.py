"""Shallow encoder on bf16 MFMA matrix cores (learner side).

Autograd glue over csrc/shallow_mfma.hip + shallow_mfma_bwd.hip.
DEFAULT path since round 2 (hardware parity + perf evidence in
profiles/r4_mfma_attribution.md); IMPALA_SHALLOW_MFMA=0 restores the
fp32 LDS-fused encoder (ops/encoder.py) for A/B.

Weight packings (pure torch reshapes, done per forward; ~20 KiB total):
  w1p [16][256]  = conv1.weight.flatten(1)            k = c*64+ky*8+kx
  w2p [32][256]  = conv2.weight.flatten(1)            k = c*16+ky*4+kx
  w3p [32][288]  = conv3.weight.permute(0,2,3,1).flatten(1)  k = tap*32+c
Backward repacks:
  wr3 [32][288]  = conv3.weight.flip(2,3).permute(1,2,3,0).flatten(1)
  wr2 [4][16][128]: class (py,px): w2[:, :, py::2, px::2]
                    .permute(1,2,3,0).flatten(1)      k = (kyi*2+kxi)*32+f
dW/db returns arrive already in the torch layouts: the fused
bwd_finalize kernel does the slab sums, layout transposes and bias
column sums in one launch (round 2).
"""
from __future__ import annotations

import os

import torch

from . import ext_hip

# IMPALA_MFMA_STASH=1: keep references to the last call's intermediates
# for post-hoc finiteness forensics (scripts/nan_probe.py --gradwatch).
# Refs only — on the graph path these alias pool tensors, valid until
# the next replay.
STASH: dict = {}


def _stash_enabled() -> bool:
    return os.environ.get("IMPALA_MFMA_STASH", "0") == "1"


def shallow_mfma_enabled() -> bool:
    """Default ON since round 2: hardware parity 9/9 PASS (r2 probe) and
    875 vs 1218 us/step of GPU kernel time against the fp32 fused chain
    (profiles/r4_mfma_attribution.md); the extra eager launches it costs
    disappear under the hipGraph-captured product step.
    IMPALA_SHALLOW_MFMA=0 restores the fp32 encoder for A/B."""
    return os.environ.get("IMPALA_SHALLOW_MFMA", "1") != "0"


class _ShallowMfmaEncode(torch.autograd.Function):
    @staticmethod
    def forward(ctx, frames, w1, b1, w2, b2, w3, b3):
        ext = ext_hip()
        bf = torch.bfloat16
        w1p = w1.detach().flatten(1).to(bf).contiguous()
        w2p = w2.detach().flatten(1).to(bf).contiguous()
        w3p = w3.detach().permute(0, 2, 3, 1).flatten(1).to(bf).contiguous()
        a1, a2, a3 = ext.shallow_mfma_fwd(
            frames.contiguous(), w1p, b1.detach().float().contiguous(),
            w2p, b2.detach().float().contiguous(), w3p,
            b3.detach().float().contiguous())
        ctx.save_for_backward(frames, a1, a2, a3, w2, w3)
        if _stash_enabled():
            STASH.update(frames=frames, a1=a1, a2=a2, a3=a3,
                         w1p=w1p, w2p=w2p, w3p=w3p)
        return a3

    @staticmethod
    def backward(ctx, d_a3):
        ext = ext_hip()
        frames, a1, a2, a3, w2, w3 = ctx.saved_tensors
        bf = torch.bfloat16
        dz3 = ext.shallow_mask_convert(d_a3.float().contiguous(), a3)
        wr3 = (w3.detach().flip(2, 3).permute(1, 2, 3, 0).flatten(1)
               .to(bf).contiguous())
        cls = []
        for py in range(2):
            for px in range(2):
                cls.append(w2.detach()[:, :, py::2, px::2]
                           .permute(1, 2, 3, 0).reshape(16, 128))
        wr2 = torch.stack(cls).to(bf).contiguous()
        (da1, da2, dw1, dw2, dw3, db1, db2, db3) = ext.shallow_mfma_bwd(
            frames.contiguous(), a1, a2, dz3, wr3, wr2)
        if _stash_enabled():
            STASH.update(d_a3=d_a3, dz3=dz3, wr3=wr3, wr2=wr2, da1=da1,
                         da2=da2, dw1=dw1, dw2=dw2, dw3=dw3, db1=db1,
                         db2=db2, db3=db3)
        # weight grads arrive already in torch layouts and the bias
        # column sums already reduced (fused bwd_finalize kernel)
        return None, dw1, db1, dw2, db2, dw3, db3


def shallow_mfma_encode(net, frames_u8: torch.Tensor) -> torch.Tensor:
    """(N,4,84,84) u8 -> (N,1568) fp32 conv features (NCHW-flat order,
    drop-in for ops.encoder.shallow_encode_conv)."""
    a3 = _ShallowMfmaEncode.apply(
        frames_u8, net.conv1.weight, net.conv1.bias, net.conv2.weight,
        net.conv2.bias, net.conv3.weight, net.conv3.bias)
    return a3.float()


def shallow_mfma_features(net, frames_u8: torch.Tensor) -> torch.Tensor:
    """Full encode incl. the fc: the conv stack on the MFMA kernels, the
    fc on the in-tree bf16 split-K MFMA GEMM (ops/linear.py,
    v_mfma_f32_16x16x32_bf16, fp32 masters).

    IMPALA_MFMA_FC: "bf16" (default) | "fp32" (the fp32 split-K GEMM) |
    "rocblas" (hipBLASLt bf16 via tracked casts — the round-2 default
    until r19/r21 forensics traced transient garbage fc-grads under
    hipGraph replay to that path; see profiles/r2_nan_rootcause.md)."""
    a3 = _ShallowMfmaEncode.apply(
        frames_u8, net.conv1.weight, net.conv1.bias, net.conv2.weight,
        net.conv2.bias, net.conv3.weight, net.conv3.bias)
    mode = os.environ.get("IMPALA_MFMA_FC", "bf16")
    if mode == "rocblas":
        bf = torch.bfloat16
        y = torch.nn.functional.linear(a3, net.fc.weight.to(bf),
                                       net.fc.bias.to(bf))
        return torch.relu(y).float()
    if mode == "fp32":
        from torched_impala_amd.ops.linear import linear_relu
        return linear_relu(net.fc, a3.float())
    from torched_impala_amd.ops.linear import linear_relu_bf16
    return linear_relu_bf16(net.fc, a3)
