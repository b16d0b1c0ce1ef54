"""Deep IMPALA ResNet encoder on the NHWC bf16 MFMA kernels.

Forward-only for now (round-2 ladder: docs/deep_conv_design.md).  The
whole 15-conv + 3-pool stack of `models.DeepResNet` runs as a chain of
`deep_conv3x3` / `deep_maxpool` calls in NHWC bf16, with weights
pre-packed once per refresh into the kernel's [cout][3][3][cin] layout.
The residual adds and the pre-conv ReLUs of the res blocks are fused
into the conv kernels, so a section is exactly 5 kernel launches + 1
pool (vs ~15 eager ops on the MIOpen NCHW path).

Kernels are hardware-validated by csrc/deep_probe.cpp
(profiles/deep_probe_r01.log); measured per-layer throughput in
profiles/deep_bench_r01.log.

Usage (inference / actor-side eval; the learner keeps the autograd
MIOpen path until the custom backward lands):

    packed = DeepEncoderWeights(model)       # after each weight pull
    feats = packed.encode(frames_u8_cuda)    # [N, 3456] fp32
"""
from __future__ import annotations

import torch

from . import ext_hip


class _DeepConv3x3(torch.autograd.Function):
    """Fused conv3x3(pad1) NHWC bf16 with optional pre-conv ReLU and
    residual add; grads to x/res (bf16 NHWC) and to the model's fp32
    [cout][cin][3][3] weight + bias.  Kernels: deep_kernels.hip,
    hardware-validated in profiles/deep_probe_bwd_r01.log."""

    @staticmethod
    def forward(ctx, x, w, b, res, relu_in):
        ext = ext_hip()
        cin = x.shape[3]
        packed = pack_conv_weight(w, cin)          # [cout][3][3][cin] bf16
        y = ext.deep_conv3x3(x.contiguous(), packed,
                             b.detach().float().contiguous(),
                             res=res.contiguous() if res is not None
                             else None,
                             relu_in=relu_in)
        ctx.relu_in = relu_in
        ctx.has_res = res is not None
        ctx.cin_model = w.shape[1]
        ctx.save_for_backward(x, packed)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = ext_hip()
        x, packed = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            # repack: wt[cin][3][3][cout] = packed rot180 + transposed
            wt = packed.flip(1, 2).permute(3, 1, 2, 0).contiguous()
            dx = ext.deep_conv3x3_dx(dy, wt,
                                     xmask=x if ctx.relu_in else None)
        if ctx.needs_input_grad[1]:
            dwk = ext.deep_conv3x3_dw(x.contiguous(), dy,
                                      relu_in=ctx.relu_in)
            dw = dwk.permute(0, 3, 1, 2)[:, :ctx.cin_model].contiguous()
        if ctx.needs_input_grad[2]:
            db = dy.float().sum(dim=(0, 1, 2))
        dres = dy if ctx.has_res else None
        return dx, dw, db, dres, None


class _DeepMaxpool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = ext_hip()
        y, am = ext.deep_maxpool(x.contiguous(), want_argmax=True)
        ctx.hw = (x.shape[1], x.shape[2])
        ctx.save_for_backward(am)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = ext_hip()
        (am,) = ctx.saved_tensors
        return ext.deep_maxpool_bwd(dy.contiguous(), am, *ctx.hw)


def deep_encode_train(model, frames_u8: torch.Tensor) -> torch.Tensor:
    """Differentiable DeepResNet.encode on the NHWC bf16 kernels:
    u8 [N][3][H][W] -> fp32 [N, 256] (sections + ReLU + fc + ReLU).
    Weight/bias grads land on the model's fp32 params; activations are
    bf16 NHWC throughout the conv stack (docs/deep_conv_design.md §5)."""
    ext = ext_hip()
    x = ext.deep_u8_to_nhwc4(frames_u8.contiguous())
    for sec in model.sections:
        x = _DeepConv3x3.apply(x, sec.conv.weight, sec.conv.bias, None,
                               False)
        x = _DeepMaxpool.apply(x)
        y = _DeepConv3x3.apply(x, sec.res0.conv0.weight, sec.res0.conv0.bias,
                               None, True)
        x = _DeepConv3x3.apply(y, sec.res0.conv1.weight, sec.res0.conv1.bias,
                               x, True)
        y = _DeepConv3x3.apply(x, sec.res1.conv0.weight, sec.res1.conv0.bias,
                               None, True)
        x = _DeepConv3x3.apply(y, sec.res1.conv1.weight, sec.res1.conv1.bias,
                               x, True)
    n = x.shape[0]
    flat = torch.relu(x.float()).permute(0, 3, 1, 2).reshape(n, -1)
    return torch.relu(torch.nn.functional.linear(
        flat, model.fc.weight, model.fc.bias))


def pack_conv_weight(w: torch.Tensor, cin_pad: int | None = None
                     ) -> torch.Tensor:
    """[cout][cin][3][3] fp32 -> contiguous [cout][3][3][cin_pad] bf16."""
    cout, cin, kh, kw = w.shape
    assert kh == 3 and kw == 3
    if cin_pad is None:
        cin_pad = cin
    out = torch.zeros(cout, 3, 3, cin_pad, dtype=torch.bfloat16,
                      device=w.device)
    out[:, :, :, :cin] = w.detach().permute(0, 2, 3, 1).to(torch.bfloat16)
    return out.contiguous()


class DeepEncoderWeights:
    """Pre-packed kernel-layout weights for one DeepResNet.

    `refresh()` re-packs from the live model (cheap: ~120 KiB total);
    call it after every weight pull on an actor/eval path.
    """

    def __init__(self, model):
        from ..models.nets import DeepResNet
        assert isinstance(model, DeepResNet), "deep encoder only"
        assert model.obs_shape[0] == 3, "u8->nhwc4 entry expects 3 channels"
        self._model = model
        self.refresh()

    def refresh(self):
        convs = []
        for i, sec in enumerate(self._model.sections):
            cin_pad = 4 if i == 0 else None
            convs.append([
                (pack_conv_weight(sec.conv.weight, cin_pad),
                 sec.conv.bias.detach().float().contiguous()),
                (pack_conv_weight(sec.res0.conv0.weight),
                 sec.res0.conv0.bias.detach().float().contiguous()),
                (pack_conv_weight(sec.res0.conv1.weight),
                 sec.res0.conv1.bias.detach().float().contiguous()),
                (pack_conv_weight(sec.res1.conv0.weight),
                 sec.res1.conv0.bias.detach().float().contiguous()),
                (pack_conv_weight(sec.res1.conv1.weight),
                 sec.res1.conv1.bias.detach().float().contiguous()),
            ])
        self._convs = convs

    def encode_nhwc(self, frames_u8: torch.Tensor) -> torch.Tensor:
        """u8 [N][3][H][W] -> bf16 NHWC [N][H'][W'][32] (post-sections,
        pre-ReLU/fc)."""
        ext = ext_hip()
        x = ext.deep_u8_to_nhwc4(frames_u8.contiguous())
        for sec_w in self._convs:
            (wh, bh), (w00, b00), (w01, b01), (w10, b10), (w11, b11) = sec_w
            x = ext.deep_conv3x3(x, wh, bh)
            x = ext.deep_maxpool(x)[0]
            y = ext.deep_conv3x3(x, w00, b00, relu_in=True)
            x = ext.deep_conv3x3(y, w01, b01, res=x, relu_in=True)
            y = ext.deep_conv3x3(x, w10, b10, relu_in=True)
            x = ext.deep_conv3x3(y, w11, b11, res=x, relu_in=True)
        return x

    @torch.no_grad()
    def encode(self, frames_u8: torch.Tensor) -> torch.Tensor:
        """u8 [N][3][H][W] -> fp32 [N, flat] features: the full
        DeepResNet.encode (sections + ReLU + fc + ReLU)."""
        m = self._model
        x = self.encode_nhwc(frames_u8)
        n = x.shape[0]
        flat = torch.relu(x.float()).permute(0, 3, 1, 2).reshape(n, -1)
        return torch.relu(torch.nn.functional.linear(
            flat, m.fc.weight, m.fc.bias))
