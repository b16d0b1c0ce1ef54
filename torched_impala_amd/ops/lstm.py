"""Fast LSTM core: the whole unroll in one HIP kernel launch per
direction (csrc/lstm_kernels.hip).

The input-to-gate GEMM for all T steps is hoisted into one rocBLAS GEMM
(Xg = x @ W_ih^T + biases); the sequential recurrence runs as a single
batch-row-partitioned persistent kernel. Weight gradients are reduced
with one big GEMM over (T*B) in backward (MFMA-shaped work stays on the
BLAS path).

Numerics: fp32 parity with nn.LSTMCell iterated eagerly (tested in
tests/test_gpu_kernels.py::test_lstm_seq_parity*).
"""
from __future__ import annotations

import os

import torch

from torched_impala_amd import ops


def lstm_bf16_weights() -> bool:
    """bf16 recurrent weights (fp32 masters + accumulation). Measured
    ~1% end-to-end (r9 A/B: 1.256 vs 1.267 ms/step) — the recurrence is
    barrier/latency-bound, not W-bandwidth-bound — so the exact-fp32
    path stays the DEFAULT; IMPALA_LSTM_BF16=1 enables the bf16 A/B."""
    return os.environ.get("IMPALA_LSTM_BF16", "0") == "1"


class _LstmSeq(torch.autograd.Function):
    @staticmethod
    def forward(ctx, Xg, W_hh, h0, c0, notdone):
        ext = ops.ext_hip()
        # NB: grad mode is disabled inside Function.forward —
        # needs_input_grad is the correct signal here.
        need_bwd = ctx.needs_input_grad[0] or ctx.needs_input_grad[1]
        bf16w = lstm_bf16_weights()
        W_hhT = W_hh.detach().t().contiguous()
        if bf16w:
            W_hhT = W_hhT.to(torch.bfloat16)
        H, act, C, Hp, hT, cT = ext.lstm_seq_fwd(
            Xg.contiguous(), W_hhT, notdone.contiguous(),
            h0.contiguous(), c0.contiguous(), need_bwd)
        ctx.bf16w = bf16w
        if need_bwd:
            ctx.save_for_backward(act, C, Hp, c0, notdone, W_hh)
        ctx.need_bwd = need_bwd
        return H, hT, cT

    @staticmethod
    def backward(ctx, dH, dhT, dcT):
        ext = ops.ext_hip()
        act, C, Hp, c0, notdone, W_hh = ctx.saved_tensors
        B = dH.shape[1]
        if dhT is None:
            dhT = torch.zeros(B, 256, device=dH.device, dtype=dH.dtype)
        if dcT is None:
            dcT = torch.zeros(B, 256, device=dH.device, dtype=dH.dtype)
        W = W_hh.detach().contiguous()
        if getattr(ctx, "bf16w", False):
            W = W.to(torch.bfloat16)
        dG, dh0, dc0 = ext.lstm_seq_bwd(
            dH.contiguous(), dhT.contiguous(), dcT.contiguous(),
            act, C, c0.contiguous(), notdone.contiguous(), W)
        # dW_hh = sum_{t,b} dG[t,b,:]^T Hp[t,b,:]  — one (1024,TB)x(TB,256) GEMM
        dW_hh = dG.reshape(-1, 1024).t().mm(Hp.reshape(-1, 256))
        return dG, dW_hh, dh0, dc0, None


def lstm_core_forward(core: torch.nn.LSTMCell, core_input: torch.Tensor,
                      done: torch.Tensor, core_state):
    """HIP fast path for _CoreMixin._run_core.

    core_input: (T, B, F); done: (T, B) bool; core_state: (h0, c0).
    Returns (core_output (T*B, 256), (hT, cT)).
    """
    T, B, F = core_input.shape
    assert core.hidden_size == 256, "HIP LSTM core is specialized for 256"
    h0, c0 = core_state
    # the recurrence kernel is fp32; keep the gate GEMM out of any
    # surrounding autocast region (which would emit bf16)
    with torch.autocast("cuda", enabled=False):
        bias = core.bias_ih + core.bias_hh
        Xg = torch.addmm(bias, core_input.reshape(T * B, F).float(),
                         core.weight_ih.t()).view(T, B, 1024)
        notdone = (~done).float()
        H, hT, cT = _LstmSeq.apply(Xg, core.weight_hh, h0.float(),
                                   c0.float(), notdone)
    return H.reshape(T * B, 256), (hT, cT)
