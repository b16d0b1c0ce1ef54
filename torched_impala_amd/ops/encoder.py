"""Fused shallow-encoder op: conv1->conv2->conv3 in one per-image kernel
(csrc/conv_kernels.hip), u8 frames in, (N, 1568) flat features out.

The backward kernel emits per-image weight-grad slabs which are reduced
with one sum and split into dW1/db1/dW2/db2/dW3/db3 (deterministic — no
atomics). No gradient flows to the frames (they are env observations).
The fc/heads GEMMs stay on the MFMA BLAS path in the model.
"""
from __future__ import annotations

import torch

from torched_impala_amd import ops

_SHAPES = dict(w1=(16, 4, 8, 8), w2=(32, 16, 4, 4), w3=(32, 32, 3, 3))
_N1 = 16 * 4 * 8 * 8
_N2 = 32 * 16 * 4 * 4
_N3 = 32 * 32 * 3 * 3


class _ShallowEncode(torch.autograd.Function):
    @staticmethod
    def forward(ctx, frames, w1, b1, w2, b2, w3, b3):
        ext = ops.ext_hip()
        need_bwd = any(ctx.needs_input_grad[1:])
        a1, a2, a3 = ext.shallow_encoder_fwd(
            frames.contiguous(), w1.contiguous(), b1.contiguous(),
            w2.contiguous(), b2.contiguous(), w3.contiguous(),
            b3.contiguous(), need_bwd)
        if need_bwd:
            ctx.save_for_backward(frames, a1, a2, a3, w2, w3)
        return a3

    @staticmethod
    def backward(ctx, d_a3):
        ext = ops.ext_hip()
        frames, a1, a2, a3, w2, w3 = ctx.saved_tensors
        slab = ext.shallow_encoder_bwd(
            frames, a1, a2, a3, d_a3.contiguous(), w2.contiguous(),
            w3.contiguous())
        g = slab.sum(dim=0)
        off = 0
        dw1 = g[off:off + _N1].view(_SHAPES["w1"]); off += _N1
        dw2 = g[off:off + _N2].view(_SHAPES["w2"]); off += _N2
        dw3 = g[off:off + _N3].view(_SHAPES["w3"]); off += _N3
        db1 = g[off:off + 16]; off += 16
        db2 = g[off:off + 32]; off += 32
        db3 = g[off:off + 32]
        return None, dw1, db1, dw2, db2, dw3, db3


def shallow_encode_conv(net, x_u8: torch.Tensor) -> torch.Tensor:
    """(N, 4, 84, 84) u8 -> (N, 1568) flat conv features via HIP."""
    return _ShallowEncode.apply(
        x_u8, net.conv1.weight, net.conv1.bias, net.conv2.weight,
        net.conv2.bias, net.conv3.weight, net.conv3.bias)
