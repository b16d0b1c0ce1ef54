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
