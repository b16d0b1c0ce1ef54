"""MFMA linear op: Y = relu(X @ W^T + b) on v_mfma_f32_16x16x4_f32
(csrc/linear_kernels.hip). Forward fuses bias+ReLU; backward-dX fuses
the ReLU mask into the A-staging pass. dW/db are rocBLAS reductions over
the big M dim (transposed layouts consumed natively)."""
from __future__ import annotations

import torch

from torched_impala_amd import ops


class _LinearReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b):
        ext = ops.ext_hip()
        y = ext.linear_mfma(x.contiguous(), w.contiguous(),
                            b.contiguous(), None, 1)
        ctx.save_for_backward(x, w, y)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = ops.ext_hip()
        x, w, y = ctx.saved_tensors
        dy = dy.contiguous()
        dx = None
        if ctx.needs_input_grad[0]:
            wt = w.detach().t().contiguous()
            dx = ext.linear_mfma(dy, wt, None, y, 2)  # mask fused
        dym = dy * (y > 0)
        dw = dym.t().mm(x) if ctx.needs_input_grad[1] else None
        db = dym.sum(0) if ctx.needs_input_grad[2] else None
        return dx, dw, db


def linear_relu(module: torch.nn.Linear, x: torch.Tensor) -> torch.Tensor:
    if not (x.is_cuda and ops.hip_available()):
        # eager fallback (CPU tests / escape hatch) — same math
        return torch.relu(torch.nn.functional.linear(
            x, module.weight, module.bias))
    return _LinearReLU.apply(x, module.weight, module.bias)


class _LinearReLUBf16(torch.autograd.Function):
    """fc behind the bf16 MFMA encoder: in-tree bf16 GEMMs for
    fwd/dX/dW (v_mfma_f32_16x16x32_bf16, split-K), fp32 master weights.
    Replaces the hipBLASLt bf16 path (profiles/r2_nan_rootcause.md)."""

    @staticmethod
    def forward(ctx, x, w, b):
        # x (M,K) bf16; w (N,K) fp32 master; b (N,) fp32
        ext = ops.ext_hip()
        wb = w.detach().to(torch.bfloat16).contiguous()
        y = ext.linear_bf16(x.contiguous(), wb,
                            b.detach().float().contiguous(), 1, False)
        ctx.save_for_backward(x, wb, y)
        return y  # fp32, relu'd

    @staticmethod
    def backward(ctx, dy):
        ext = ops.ext_hip()
        x, wb, y = ctx.saved_tensors
        dym = dy * (y > 0)
        dymb = dym.to(torch.bfloat16).contiguous()
        dx = None
        if ctx.needs_input_grad[0]:
            dx = ext.linear_bf16(dymb, wb.t().contiguous(), None, 0, True)
        dw = None
        if ctx.needs_input_grad[1]:
            dw = ext.linear_bf16(dymb.t().contiguous(),
                                 x.detach().t().contiguous(), None, 0,
                                 False)
        db = dym.sum(0) if ctx.needs_input_grad[2] else None
        return dx, dw, db


def linear_relu_bf16(module: torch.nn.Linear, x: torch.Tensor
                     ) -> torch.Tensor:
    """x bf16 -> relu(fc(x)) fp32; weight grads on the fp32 masters."""
    if not (x.is_cuda and ops.hip_available()):
        return torch.relu(torch.nn.functional.linear(
            x.float(), module.weight, module.bias))
    return _LinearReLUBf16.apply(x, module.weight, module.bias)
