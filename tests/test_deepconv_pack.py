"""CPU-side checks for the deep-encoder weight packing (ops/deepconv.py).
The GPU compose/parity tests live in test_gpu_deep.py."""
import torch

from torched_impala_amd.ops.deepconv import pack_conv_weight


def test_pack_layout():
    w = torch.arange(2 * 3 * 3 * 3, dtype=torch.float32).view(2, 3, 3, 3)
    p = pack_conv_weight(w)
    assert p.shape == (2, 3, 3, 3) and p.dtype == torch.bfloat16
    # [cout][cin][ky][kx] -> [cout][ky][kx][cin]
    for f in range(2):
        for ky in range(3):
            for kx in range(3):
                for c in range(3):
                    assert p[f, ky, kx, c].float() == float(
                        w[f, c, ky, kx].to(torch.bfloat16))


def test_pack_pads_cin():
    w = torch.randn(16, 3, 3, 3)
    p = pack_conv_weight(w, 4)
    assert p.shape == (16, 3, 3, 4)
    assert torch.all(p[..., 3] == 0)
    assert torch.equal(p[..., :3].float(),
                       w.permute(0, 2, 3, 1).to(torch.bfloat16).float())


def test_deep_encoder_weights_requires_deep_model():
    import pytest
    from torched_impala_amd.models import ShallowNet
    from torched_impala_amd.ops.deepconv import DeepEncoderWeights
    with pytest.raises(AssertionError):
        DeepEncoderWeights(ShallowNet((4, 84, 84), 6))
