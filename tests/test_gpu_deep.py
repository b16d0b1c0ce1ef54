"""Deep ResNet NHWC bf16 MFMA kernel glue parity (design
docs/deep_conv_design.md).  The kernels themselves are hardware-validated
by the torch-free csrc/deep_probe.cpp (profiles/deep_probe_r01.log);
these tests cover the torch extension plumbing and the full-encoder
composition against the eager fp32 DeepResNet."""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

from torched_impala_amd import ops
from torched_impala_amd.models.nets import DeepResNet
from torched_impala_amd.ops.deepconv import DeepEncoderWeights, \
    pack_conv_weight


def _dev():
    return torch.device("cuda", 0)


def _rel_err(got, want):
    return ((got.float() - want).abs().max() /
            want.abs().max().clamp_min(1.0)).item()


@pytest.mark.parametrize("cin,cout,hw,relu_in,res",
                         [(16, 16, (36, 48), True, True),
                          (16, 32, (36, 48), False, False),
                          (32, 32, (18, 24), True, True),
                          (32, 32, (9, 12), True, False)])
def test_deep_conv3x3_parity(cin, cout, hw, relu_in, res):
    torch.manual_seed(0)
    ext = ops.ext_hip()
    H, W = hw
    N = 5
    x = torch.randn(N, cin, H, W, device=_dev())
    w = torch.randn(cout, cin, 3, 3, device=_dev()) * 0.2
    b = torch.randn(cout, device=_dev())
    r = torch.randn(N, cout, H, W, device=_dev())

    xin = F.relu(x) if relu_in else x
    want = F.conv2d(xin.to(torch.bfloat16).float(), w, b, padding=1)
    if res:
        want = want + r.to(torch.bfloat16).float()

    x_nhwc = x.permute(0, 2, 3, 1).to(torch.bfloat16).contiguous()
    r_nhwc = r.permute(0, 2, 3, 1).to(torch.bfloat16).contiguous()
    got = ext.deep_conv3x3(x_nhwc, pack_conv_weight(w), b,
                           res=r_nhwc if res else None, relu_in=relu_in)
    got = got.permute(0, 3, 1, 2)
    assert _rel_err(got, want) < 2e-2


def test_deep_conv3x3_cin4_pad():
    """conv1: 3 real channels zero-padded to 4 must equal the 3-ch conv."""
    torch.manual_seed(1)
    ext = ops.ext_hip()
    N, H, W = 3, 72, 96
    x = torch.randn(N, 3, H, W, device=_dev())
    w = torch.randn(16, 3, 3, 3, device=_dev()) * 0.2
    b = torch.randn(16, device=_dev())
    want = F.conv2d(x.to(torch.bfloat16).float(), w, b, padding=1)
    x4 = torch.zeros(N, H, W, 4, dtype=torch.bfloat16, device=_dev())
    x4[..., :3] = x.permute(0, 2, 3, 1).to(torch.bfloat16)
    got = ext.deep_conv3x3(x4.contiguous(), pack_conv_weight(w, 4), b)
    assert _rel_err(got.permute(0, 3, 1, 2), want) < 2e-2


def test_deep_maxpool_parity():
    torch.manual_seed(2)
    ext = ops.ext_hip()
    x = torch.randn(4, 16, 36, 48, device=_dev()).to(torch.bfloat16)
    want = F.max_pool2d(x.float(), 3, stride=2, padding=1)
    got = ext.deep_maxpool(x.permute(0, 2, 3, 1).contiguous())
    assert torch.equal(got.permute(0, 3, 1, 2).float(), want)


def test_deep_u8_entry_parity():
    ext = ops.ext_hip()
    g = torch.Generator(device="cpu").manual_seed(3)
    frames = torch.randint(0, 256, (4, 3, 72, 96), dtype=torch.uint8,
                           generator=g).to(_dev())
    got = ext.deep_u8_to_nhwc4(frames)
    want = (frames.float() / 255.0).to(torch.bfloat16)
    assert torch.equal(got[..., :3].permute(0, 3, 1, 2), want)
    assert torch.all(got[..., 3] == 0)


def test_deep_encoder_full_stack_parity():
    """The whole 15-conv/3-pool encoder vs eager fp32 DeepResNet.encode.
    bf16 accumulates error over 6 convs per path; tolerance is the bf16
    train-quality bound from the design note (§5)."""
    torch.manual_seed(4)
    model = DeepResNet((3, 72, 96), 15).to(_dev()).eval()
    enc = DeepEncoderWeights(model)
    g = torch.Generator(device="cpu").manual_seed(5)
    frames = torch.randint(0, 256, (8, 3, 72, 96), dtype=torch.uint8,
                           generator=g).to(_dev())
    with torch.no_grad():
        want = model.encode(frames.float())
    got = enc.encode(frames)
    assert got.shape == want.shape
    err = _rel_err(got, want)
    assert err < 5e-2, err


def test_deep_encoder_refresh_tracks_weights():
    torch.manual_seed(6)
    model = DeepResNet((3, 72, 96), 15).to(_dev()).eval()
    enc = DeepEncoderWeights(model)
    frames = torch.randint(0, 256, (2, 3, 72, 96), dtype=torch.uint8
                           ).to(_dev())
    a = enc.encode(frames)
    with torch.no_grad():
        for p in model.parameters():
            p.mul_(0.5)
    enc.refresh()
    b = enc.encode(frames)
    assert not torch.allclose(a, b)
