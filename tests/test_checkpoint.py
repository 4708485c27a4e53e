"""Per-stage checkpoint save/load (the reference's weight dispatch,
dispatcher.py:57, as a disk artifact)."""

import torch

from defer_amd.checkpoint import load_manifest, load_stage, save_stages
from defer_amd.models import resnet50
from defer_amd.parallel.partitioner import partition_model


def test_stage_checkpoint_roundtrip(tmp_path):
    m = resnet50()
    cuts = ["add_4", "add_12"]
    d = str(tmp_path / "ckpt")
    save_stages(m, cuts, d)
    man = load_manifest(d)
    assert man["num_stages"] == 3 and man["cut_points"] == cuts

    x = torch.randn(1, 64, 64, 3)
    with torch.no_grad():
        want = m(x)

    torch.manual_seed(123)  # fresh (different) weights
    m2 = resnet50()
    stages = partition_model(m2, cuts)
    for i, s in enumerate(stages):
        load_stage(s, d, i)
    with torch.no_grad():
        z = x
        for s in stages:
            z = s(z)
    assert torch.equal(z, want)
