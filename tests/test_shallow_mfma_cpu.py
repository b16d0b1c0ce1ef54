"""CPU verification of the shallow MFMA autograd glue
(ops/shallow_mfma.py) with a torch-eager mock of the extension — the
kernels' layouts and semantics re-implemented with conv2d/conv grads so
every weight pack, backward repack and dW reshape in the python glue is
checked deterministically (the kernels themselves are covered by
deep_probe sections 8-9 on hardware)."""
import pytest
import torch
import torch.nn.functional as F

from torched_impala_amd.models.nets import ShallowNet
from torched_impala_amd.ops import shallow_mfma as S

BF = torch.bfloat16


def _q(x):
    return x.to(BF).float()


class _MockExt:
    """torch-eager mock with the kernel layouts: a1/a2 NHWC bf16,
    a3 NCHW-flat bf16, k-major dW slabs."""

    @staticmethod
    def shallow_mfma_fwd(frames, w1p, b1, w2p, b2, w3p, b3):
        n = frames.shape[0]
        x = _q((frames.float() / 255.0))
        w1 = w1p.float().view(16, 4, 8, 8)
        a1 = F.relu(F.conv2d(x, w1, b1, stride=4))            # [n,16,20,20]
        a1 = _q(a1)
        w2 = w2p.float().view(32, 16, 4, 4)
        a2 = F.relu(F.conv2d(a1, w2, b2, stride=2))           # [n,32,9,9]
        a2 = _q(a2)
        w3 = w3p.float().view(32, 9, 32).permute(0, 2, 1).reshape(
            32, 32, 3, 3)                                     # k=tap*32+c
        a3 = F.relu(F.conv2d(a2, w3, b3))                     # [n,32,7,7]
        return [a1.permute(0, 2, 3, 1).to(BF).contiguous(),
                a2.permute(0, 2, 3, 1).to(BF).contiguous(),
                a3.reshape(n, 1568).to(BF).contiguous()]

    @staticmethod
    def shallow_mask_convert(d, act):
        return torch.where(act.float() > 0, d, torch.zeros(())).to(BF)

    @staticmethod
    def shallow_mfma_bwd(frames, a1, a2, dz3, wr3, wr2):
        n = frames.shape[0]
        a1n = a1.float().permute(0, 3, 1, 2)                  # NCHW
        a2n = a2.float().permute(0, 3, 1, 2)
        dz3n = dz3.float().view(n, 32, 7, 7)
        # conv3 dX via the repacked wr3 [cin][288] (k = tap*32 + f):
        # full correlation = conv2d of 2-padded dz3 with wr3 as weights
        w3r = wr3.float().view(32, 9, 32).permute(0, 2, 1).reshape(
            32, 32, 3, 3)                                     # [cin][f][ky][kx]
        da2 = F.conv2d(F.pad(dz3n, (2, 2, 2, 2)), w3r)
        da2 = da2 * (a2n > 0)
        da2 = _q(da2)
        # conv2 dX via torch grad.conv2d_input from wr2-equivalent w2
        w2 = _MockExt._w2_from_wr2(wr2)
        da1 = torch.nn.grad.conv2d_input(
            (n, 16, 20, 20), w2, _q(da2), stride=2)
        da1 = da1 * (a1n > 0)
        da1 = _q(da1)
        # dW (k-major) from the bf16-rounded operands
        x = _q(frames.float() / 255.0)
        dw1 = torch.nn.grad.conv2d_weight(
            x, (16, 4, 8, 8), _q(da1), stride=4)
        dw2 = torch.nn.grad.conv2d_weight(
            a1n, (32, 16, 4, 4), _q(da2), stride=2)
        dw3 = torch.nn.grad.conv2d_weight(a2n, (32, 32, 3, 3), dz3n)
        da1b = da1.permute(0, 2, 3, 1).to(BF).contiguous()
        da2b = da2.permute(0, 2, 3, 1).to(BF).contiguous()
        # round-2 contract: torch-layout dW + fused bias column sums
        # (bf16-rounded da*, matching the device bwd_finalize inputs)
        db1 = da1b.float().sum(dim=(0, 1, 2))
        db2 = da2b.float().sum(dim=(0, 1, 2))
        db3 = dz3n.sum(dim=(0, 2, 3))
        return [da1b, da2b, dw1.contiguous(), dw2.contiguous(),
                dw3.contiguous(), db1, db2, db3]

    @staticmethod
    def _w2_from_wr2(wr2):
        # wr2 [4][16][128]: cls=(py,px), k=(kyi*2+kxi)*32+f -> w2 torch
        w2 = torch.zeros(32, 16, 4, 4)
        v = wr2.float().view(4, 16, 4, 32)       # [cls][c][t][f]
        for cls in range(4):
            py, px = cls >> 1, cls & 1
            for t in range(4):
                kyi, kxi = t >> 1, t & 1
                w2[:, :, py + 2 * kyi, px + 2 * kxi] = v[cls, :, t].T
        return w2


@pytest.fixture()
def mock_ext(monkeypatch):
    monkeypatch.setattr(S, "ext_hip", lambda: _MockExt)
    return _MockExt


def _setup(seed):
    torch.manual_seed(seed)
    net = ShallowNet((4, 84, 84), 6)
    g = torch.Generator().manual_seed(seed + 1)
    frames = torch.randint(0, 256, (3, 4, 84, 84), dtype=torch.uint8,
                           generator=g)
    return net, frames


def test_mock_forward_matches_eager(mock_ext):
    net, frames = _setup(30)
    with torch.no_grad():
        x = frames.float() / 255.0
        want = F.relu(net.conv3(F.relu(net.conv2(F.relu(net.conv1(x))))))
        want = want.flatten(1)
        got = S.shallow_mfma_encode(net, frames)
    rel = (got - want).abs().max() / want.abs().max().clamp_min(1.0)
    assert rel.item() < 5e-2, rel.item()


def test_mock_grads_match_eager(mock_ext):
    net, frames = _setup(31)
    tgt = torch.randn(3, 1568)

    def grads(use_mock):
        net.zero_grad(set_to_none=True)
        if use_mock:
            out = S.shallow_mfma_encode(net, frames)
        else:
            x = frames.float() / 255.0
            out = F.relu(net.conv3(F.relu(net.conv2(
                F.relu(net.conv1(x)))))).flatten(1)
        ((out - tgt) ** 2).mean().backward()
        return {k: p.grad.detach().clone()
                for k, p in net.named_parameters()
                if p.grad is not None and "conv" in k}

    ref = grads(False)
    got = grads(True)
    assert set(got) == set(ref)
    for k in sorted(ref):
        a, b = got[k].flatten(), ref[k].flatten()
        cos = F.cosine_similarity(a, b, dim=0).item()
        ratio = (a.norm() / b.norm().clamp_min(1e-12)).item()
        assert cos > 0.97, f"{k}: cosine {cos}"
        assert 0.8 < ratio < 1.25, f"{k}: norm ratio {ratio}"


def test_mock_features_with_fc_grads(mock_ext):
    """shallow_mfma_features: conv stack via the Function + fc on bf16
    torch autograd — fc weight grads must land on the fp32 masters."""
    net, frames = _setup(32)
    tgt = torch.randn(3, 256)
    net.zero_grad(set_to_none=True)
    out = S.shallow_mfma_features(net, frames)
    ((out - tgt) ** 2).mean().backward()
    assert net.fc.weight.grad is not None
    assert net.fc.weight.grad.dtype == torch.float32
    assert net.conv1.weight.grad is not None

    net.zero_grad(set_to_none=True)
    x = frames.float() / 255.0
    ref = F.relu(net.fc(F.relu(net.conv3(F.relu(net.conv2(
        F.relu(net.conv1(x)))))).flatten(1)))
    ((ref - tgt) ** 2).mean().backward()
    g_ref = net.fc.weight.grad.clone()
    net.zero_grad(set_to_none=True)
    out2 = S.shallow_mfma_features(net, frames)
    ((out2 - tgt) ** 2).mean().backward()
    g = net.fc.weight.grad
    cos = F.cosine_similarity(g.flatten(), g_ref.flatten(), dim=0).item()
    assert cos > 0.97, cos


def test_flag_gate_defaults_on(monkeypatch):
    """Default flipped ON in round 2 after hardware parity + the
    875-vs-1218 us/step attribution (profiles/r4_mfma_attribution.md);
    =0 is the A/B escape hatch."""
    monkeypatch.delenv("IMPALA_SHALLOW_MFMA", raising=False)
    assert S.shallow_mfma_enabled()
    monkeypatch.setenv("IMPALA_SHALLOW_MFMA", "0")
    assert not S.shallow_mfma_enabled()
    monkeypatch.setenv("IMPALA_SHALLOW_MFMA", "1")
    assert S.shallow_mfma_enabled()
