"""Architecture parity with the reference models: parameter counts of
our builders must track the reference's (read from /root/reference at
test time; skipped if the reference tree is absent)."""

import os
import sys

import pytest
import torch

REF = "/root/reference/src"
pytestmark = pytest.mark.skipif(
    not os.path.isdir(REF), reason="reference tree not mounted"
)


def _ref_models():
    sys.path.insert(0, REF)
    try:
        import models.resnet as ref_resnet
        import models.amoebanet as ref_amoeba
    finally:
        sys.path.remove(REF)
    return ref_resnet, ref_amoeba


def _params(m):
    return sum(p.numel() for p in m.parameters())


def test_amoebanet_param_parity():
    ref_resnet, ref_amoeba = _ref_models()
    from mpi4dl_amd.models.amoebanet import amoebanetd

    torch.manual_seed(0)
    ref = ref_amoeba.amoebanetd(num_classes=100, num_layers=6, num_filters=64)
    ours = amoebanetd(100, 6, 64)
    assert len(ref) == len(ours)
    assert _params(ref) == _params(ours), (
        _params(ref), _params(ours))  # exact: same genotype, same widths


@pytest.mark.parametrize("ver,n", [(1, 2), (2, 3)])
def test_resnet_param_parity(ver, n):
    ref_resnet, _ = _ref_models()
    import mpi4dl_amd.models.resnet as ours_r

    depth = (6 if ver == 1 else 9) * n + 2
    ref = getattr(ref_resnet, f"get_resnet_v{ver}")(
        (2, 3, 64, 64), depth, num_classes=10
    )
    ours = getattr(ours_r, f"get_resnet_v{ver}")((2, 3, 64, 64), 10, n=n)
    assert len(ref) == len(ours)
    rp, op = _params(ref), _params(ours)
    # small residual diff: our GAP head vs the reference's flatten head
    assert abs(rp - op) / rp < 0.02, (rp, op)
