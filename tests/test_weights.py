import torch

from torched_impala_amd.models import MLPPolicy
from torched_impala_amd.transport import WeightMirror


def test_mirror_roundtrip():
    src = MLPPolicy((4,), 2)
    dst = MLPPolicy((4,), 2)
    mirror = WeightMirror(src)
    v = mirror.pull_into(dst)
    for a, b in zip(src.parameters(), dst.parameters()):
        assert torch.equal(a, b)
    # mutate + publish + pull
    with torch.no_grad():
        for p in src.parameters():
            p.add_(1.0)
    mirror.publish(src)
    v2 = mirror.pull_into(dst, last_seen=v)
    assert v2 > v
    for a, b in zip(src.parameters(), dst.parameters()):
        assert torch.equal(a, b)
    # no new version -> no copy needed
    assert mirror.pull_into(dst, last_seen=v2) == v2
