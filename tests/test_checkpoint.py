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


def test_dist_pipeline_loads_stage_checkpoints(tmp_path):
    """PipelineConfig.weights_dir: each rank loads only its own stage's
    weights (the reference's per-node weight shipping,
    dispatcher.py:57). A fresh differently-seeded model under the
    pipeline must reproduce the checkpointed model's outputs."""
    import os

    import torch.distributed as dist

    from defer_amd import checkpoint
    from defer_amd.config import PipelineConfig
    from defer_amd.models import resnet50
    from defer_amd.parallel.pipeline import DistPipeline

    torch.manual_seed(0)
    src = resnet50()
    checkpoint.save_stages(src, [], str(tmp_path))

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29787")
        dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        torch.manual_seed(99)            # different weights
        fresh = resnet50()
        cfg = PipelineConfig(device="cpu", dtype="fp32",
                             weights_dir=str(tmp_path), backend="gloo")
        pipe = DistPipeline(fresh, cfg, (1, 64, 64, 3))
        torch.manual_seed(5)
        x = torch.randn(1, 64, 64, 3)
        got = {}
        pipe.run(1, feed=lambda k: x,
                 collect=lambda k, y: got.__setitem__(k, y.clone()))
        with torch.no_grad():
            want = src(x)
        assert torch.allclose(got[0], want, atol=1e-5), \
            (got[0] - want).abs().max()
    finally:
        dist.destroy_process_group()


def test_weights_dir_stage_count_mismatch_raises(tmp_path):
    import os

    import pytest
    import torch.distributed as dist

    from defer_amd import checkpoint
    from defer_amd.config import PipelineConfig
    from defer_amd.models import resnet50
    from defer_amd.parallel.pipeline import DistPipeline

    checkpoint.save_stages(resnet50(), ["add_8"], str(tmp_path))  # 2 stages
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29788")
        dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        cfg = PipelineConfig(device="cpu", dtype="fp32",
                             weights_dir=str(tmp_path), backend="gloo")
        with pytest.raises(ValueError, match="stages"):
            DistPipeline(resnet50(), cfg, (1, 64, 64, 3))
    finally:
        dist.destroy_process_group()


def _ckpt_worker(rank, world, port, ckpt_dir, q):
    import os

    import torch.distributed as dist

    from defer_amd.config import PipelineConfig
    from defer_amd.models import resnet50
    from defer_amd.parallel.pipeline import DistPipeline

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(50 + rank)       # ranks start from DIFFERENT
        fresh = resnet50()                 # random weights on purpose
        cfg = PipelineConfig(device="cpu", dtype="fp32",
                             weights_dir=ckpt_dir, backend="gloo")
        pipe = DistPipeline(fresh, cfg, (1, 64, 64, 3))
        assert pipe.cuts == ["add_8"]      # cuts came from the manifest
        torch.manual_seed(5)
        x = torch.randn(1, 64, 64, 3)
        got = {}
        pipe.run(1, feed=lambda k: x,
                 collect=lambda k, y: got.__setitem__(k, y.clone()))
        if rank == 0:
            torch.manual_seed(0)
            src = resnet50()
            with torch.no_grad():
                want = src(x)
            err = (got[0] - want).abs().max().item()
            q.put(("err", err))
    finally:
        dist.destroy_process_group()


def test_weights_dir_world2_manifest_cuts(tmp_path):
    """World 2 + weights_dir: cuts come from the checkpoint manifest and
    each rank loads its own part{i}.pt — even though the two ranks
    constructed models with different random seeds, the pipeline output
    matches the checkpointed model."""
    import torch.multiprocessing as mp

    from defer_amd import checkpoint
    from defer_amd.models import resnet50

    torch.manual_seed(0)
    checkpoint.save_stages(resnet50(), ["add_8"], str(tmp_path))
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_ckpt_worker,
                         args=(r, 2, 29661, str(tmp_path), q))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    assert all(p.exitcode == 0 for p in procs)
    tag, err = q.get()
    assert tag == "err" and err <= 1e-5, err


def test_defer_orchestrator_loads_checkpoints(tmp_path):
    """The threaded DEFER orchestrator honors weights_dir too: a fresh
    model on two CPU nodes reproduces the checkpointed model."""
    import queue
    import threading

    from defer_amd import DEFER, PipelineConfig, checkpoint
    from defer_amd.models import resnet50

    torch.manual_seed(0)
    src = resnet50()
    cuts = ["add_8"]
    checkpoint.save_stages(src, cuts, str(tmp_path))

    torch.manual_seed(77)
    fresh = resnet50()
    eng = DEFER(["cpu", "cpu"], config=PipelineConfig(
        device="cpu", dtype="fp32", weights_dir=str(tmp_path)))
    in_q, out_q = queue.Queue(4), queue.Queue(4)
    t = threading.Thread(target=eng.run_defer,
                         args=(fresh, cuts, in_q, out_q))
    t.start()
    torch.manual_seed(5)
    x = torch.randn(1, 64, 64, 3)
    in_q.put(x)
    in_q.put(None)
    got = out_q.get(timeout=120)
    t.join(timeout=120)
    with torch.no_grad():
        want = src(x)
    assert torch.allclose(got, want, atol=1e-5)
