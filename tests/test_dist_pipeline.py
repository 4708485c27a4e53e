"""Distributed pipeline over gloo (CPU, world_size 2/3) — the same code
path bench.py drives with RCCL on 8x MI355X (SURVEY.md §2.3 relay table).
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from defer_amd.config import PipelineConfig
from defer_amd.models import resnet50


def _worker(rank, world, port, q, cuts, steps, compression,
            dual_rail=False):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)
        from defer_amd.parallel.pipeline import DistPipeline

        model = resnet50()
        cfg = PipelineConfig(device="cpu", dtype="fp32",
                             partition_layers=cuts, ring_depth=2,
                             compression=compression, zfp_rate_bits=14,
                             dual_rail=dual_rail, backend="gloo")
        B = 1
        pipe = DistPipeline(model, cfg, (B, 64, 64, 3))

        torch.manual_seed(1234)
        inputs = [torch.randn(B, 64, 64, 3) for _ in range(steps)]
        results = {}

        pipe.run(steps, feed=lambda k: inputs[k],
                 collect=lambda k, y: results.__setitem__(k, y.clone()))

        if rank == 0:
            if compression == "fp8":
                # fp8 is a per-value cast: simulate the exact pipeline
                # (fused stages + wire quantization per hop) and demand
                # a near-bitwise match — a random-init net's softmax is
                # chaotically sensitive, so comparing against the
                # UNQUANTIZED model is meaningless for coarse codecs.
                from defer_amd.parallel.comm import Codec
                from defer_amd.parallel.partitioner import partition_model
                from defer_amd.parallel.pipeline import StageExecutor

                stages = partition_model(model, cuts)
                execs = [StageExecutor(s, "cpu", torch.float32)
                         for s in stages]

                def fwd(x):
                    z = x
                    for i, ex in enumerate(execs):
                        with torch.no_grad():
                            z = ex.run(z)
                        if i < len(execs) - 1:
                            c = Codec(cfg, tuple(z.shape),
                                      torch.float32, "cpu")
                            z = c.decode(c.encode(z, out=c.alloc_wire()))
                    return z

                want = [fwd(x) for x in inputs]
            else:
                with torch.no_grad():
                    want = [model(x) for x in inputs]
            for k in range(steps):
                assert results[k].shape == want[k].shape
                err = (results[k] - want[k]).abs().max().item()
                q.put(("err", k, err))
            q.put(("done", rank, None))
    finally:
        dist.destroy_process_group()


def _run(world, cuts, steps=4, compression="none", tol=0.0,
         dual_rail=False):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = (29501 + world + (17 if dual_rail else 0)
            + {"none": 0, "fp8": 23, "zfp": 7,
               "zfp+lz4": 11, "auto": 31}[compression])
    procs = [ctx.Process(target=_worker,
                         args=(r, world, port, q, cuts, steps, compression,
                               dual_rail))
             for r in range(world)]
    for p in procs:
        p.start()
    msgs = []
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0, f"worker failed: {p.exitcode}"
    while not q.empty():
        msgs.append(q.get())
    errs = [m[2] for m in msgs if m[0] == "err"]
    assert len(errs) == steps
    for e in errs:
        assert e <= tol, f"pipeline output mismatch: {e}"


def test_dist_pipeline_two_stage():
    _run(2, ["add_8"])


def test_dist_pipeline_three_stage_unbalanced():
    _run(3, ["add_4", "add_12"], steps=5)


@pytest.mark.slow
def test_dist_pipeline_auto_cuts():
    _run(2, None)


def test_dist_pipeline_with_zfp_compression():
    """Config 4 analogue on CPU: 2-stage pipeline with the ZFP relay.
    Lossy codec -> compare against the uncompressed model with tolerance.
    """
    _run(2, ["add_8"], steps=2, compression="zfp", tol=0.05)


def test_dist_pipeline_with_zfp_lz4_compression():
    """Full reference wire codec lz4(zfp(x)) (dispatcher.py:81-84) over a
    variable-size hop (size message + exact payload, the RCCL analogue of
    the reference's length-prefixed framing). LZ4 is lossless on top of
    ZFP, so the tolerance equals the ZFP-only case."""
    _run(2, ["add_8"], steps=3, compression="zfp+lz4", tol=0.05)


def test_node_cli_single_rank(tmp_path):
    """The node.py-style stage-worker entry runs end to end on CPU
    (reference node script parity, node.py:126-127)."""
    import subprocess
    import sys

    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29779",
               WORLD_SIZE="1", RANK="0", LOCAL_RANK="0")
    out = subprocess.run(
        [sys.executable, "-m", "defer_amd.node", "--device", "cpu",
         "--items", "2", "--batch", "2", "--report", "1"],
        capture_output=True, text=True, timeout=300, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 0, out.stderr[-2000:]
    assert "images/sec" in out.stdout


def test_eight_stage_auto_cuts_gloo():
    """The headline 8-stage chain topology (7 relay hops
    + result return), auto-partitioned, world_size 8 on gloo."""
    _run(8, None, steps=3)


def test_bench_torchrun_two_ranks_cpu():
    """The driver's exact launch form (torch.distributed.run, nnodes=1,
    loopback rendezvous) runs bench.py end to end on CPU with 2 ranks
    and emits exactly one JSON line on stdout."""
    import json
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29791", os.path.join(root, "bench.py"),
         "--gpus", "2", "--device", "cpu", "--batch", "1",
         "--steps", "4", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, cwd=root)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.strip()]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2 and d["config"]["parallelism"] == "pp2"
    assert d["value"] > 0
    # full driver-contract schema: every required field, right types
    schema = {"metric": str, "value": float, "unit": str, "n_gpus": int,
              "steps": int, "warmup": int, "ms_per_step": float,
              "higher_is_better": bool, "scaling": str,
              "dtype": str, "data": str, "config": dict}
    for key, typ in schema.items():
        assert key in d and isinstance(d[key], typ), (key, d.get(key))
    assert "vs_baseline" in d            # null allowed
    assert d["scaling"] in ("weak", "strong")
    assert d["data"] == "synthetic" and d["higher_is_better"] is True
    for key in ("model", "global_batch", "parallelism"):
        assert key in d["config"], key


def _worker_two_runs(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)
        from defer_amd.parallel.pipeline import DistPipeline

        cfg = PipelineConfig(device="cpu", dtype="fp32", ring_depth=4,
                             backend="gloo")
        pipe = DistPipeline(resnet50(), cfg, (1, 64, 64, 3))
        torch.manual_seed(7)
        xs = [torch.randn(1, 64, 64, 3) for _ in range(6)]
        got = {}
        # warmup-then-timed shape: MULTIPLE run() calls on one pipeline
        # (regression: drained send Works were re-waited and deadlocked)
        pipe.run(2, feed=lambda k: xs[k],
                 collect=lambda k, y: got.__setitem__(k, y.clone()))
        dist.barrier()
        pipe.run(4, feed=lambda k: xs[2 + k],
                 collect=lambda k, y: got.__setitem__(2 + k,
                                                      y.clone()))
        if rank == 0:
            assert sorted(got) == list(range(6))
            q.put(("done", rank, None))
    finally:
        dist.destroy_process_group()


def test_two_sequential_runs_world2():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker_two_runs, args=(r, 2, 29641, q))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=420)
    alive = [p.is_alive() for p in procs]
    for p in procs:
        if p.is_alive():
            p.terminate()
    assert not any(alive), "second run() deadlocked"
    assert all(p.exitcode == 0 for p in procs)


def test_dual_rail_four_stage():
    """Dual-rail relay at world 4: every data hop splits across a direct
    rail and a forwarder rank (hop i via rank i+2, last hop via rank 0);
    outputs must be bitwise-identical to the single-rail pipeline."""
    _run(4, ["add_4", "add_8", "add_12"], steps=6, dual_rail=True)


def test_dual_rail_world3_mixed():
    """World 3: hop 0 is dual-rail (via rank 2), hop 1 stays single-rail
    (the wrap would collide on the 0->2 link) — mixed-mode chain."""
    _run(3, ["add_4", "add_12"], steps=5, dual_rail=True)


def test_dual_rail_eight_stage_auto():
    """The headline topology with dual-rail on: 8 stages, auto cuts,
    7 dual hops + result return."""
    _run(8, None, steps=4, dual_rail=True)


def test_dual_rail_with_zfp():
    """Dual-rail over fixed-size ZFP wire (both halves of the compressed
    buffer ride different rails)."""
    _run(4, ["add_4", "add_8", "add_12"], steps=3, compression="zfp",
         tol=0.05, dual_rail=True)


def _worker_two_runs_dual(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)
        from defer_amd.parallel.pipeline import DistPipeline

        cfg = PipelineConfig(device="cpu", dtype="fp32", ring_depth=2,
                             partition_layers=["add_4", "add_8",
                                               "add_12"],
                             dual_rail=True, backend="gloo")
        pipe = DistPipeline(resnet50(), cfg, (1, 64, 64, 3))
        torch.manual_seed(7)
        xs = [torch.randn(1, 64, 64, 3) for _ in range(6)]
        got = {}
        # warmup-then-timed shape: forwarder must begin/drain cleanly
        # across run() calls
        pipe.run(2, feed=lambda k: xs[k],
                 collect=lambda k, y: got.__setitem__(k, y.clone()))
        dist.barrier()
        pipe.run(4, feed=lambda k: xs[2 + k],
                 collect=lambda k, y: got.__setitem__(2 + k, y.clone()))
        if rank == 0:
            assert sorted(got) == list(range(6))
            q.put(("done", rank, None))
    finally:
        dist.destroy_process_group()


def test_two_sequential_runs_dual_rail_world4():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker_two_runs_dual,
                         args=(r, 4, 29653, q)) for r in range(4)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=420)
    alive = [p.is_alive() for p in procs]
    for p in procs:
        if p.is_alive():
            p.terminate()
    assert not any(alive), "dual-rail second run() hung"
    assert all(p.exitcode == 0 for p in procs)


def test_bench_torchrun_dual_rail_cpu():
    """The round-2 GPU validation form on CPU: driver-style torchrun
    launch of bench.py --dual-rail with 4 ranks."""
    import json
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", "29793", os.path.join(root, "bench.py"),
         "--gpus", "4", "--device", "cpu", "--batch", "1",
         "--steps", "3", "--warmup", "1", "--dual-rail",
         "--cuts", "add_4,add_8,add_12"],
        capture_output=True, text=True, timeout=600, cwd=root)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.strip()]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    assert d["config"]["dual_rail"] is True
    assert d["value"] > 0


def test_dist_pipeline_with_fp8_wire():
    """Cast-only fp8 e4m3 wire (1 B/value): the cheap lossy codec for
    hops above xGMI line rate. Compared spec-level against the locally
    simulated quantized-boundary chain (near-bitwise)."""
    _run(2, ["add_8"], steps=2, compression="fp8", tol=1e-6)


def test_dual_rail_with_fp8_wire():
    """fp8 wire + dual-rail compose (both halves of the 1 B/value wire
    ride separate rails)."""
    _run(4, ["add_4", "add_8", "add_12"], steps=3, compression="fp8",
         tol=1e-6, dual_rail=True)


# ---------------------------------------------------------------------------
# deferred variable-size interleave (the RCCL-order protocol, forced on gloo)
# ---------------------------------------------------------------------------

def _defer_worker(rank, world, port, q, steps):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    # force the CUDA-order interleave (s_0, s_1, p_0, s_2, p_1, ...) on
    # gloo; safe only without the rank0 result-return cycle, so
    # return_results=False and the last rank collects
    os.environ["DEFER_AMD_VARRING_DEFER"] = "1"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)
        from defer_amd.parallel.pipeline import DistPipeline

        model = resnet50()
        cuts = ["add_4", "add_12"]
        cfg = PipelineConfig(device="cpu", dtype="fp32",
                             partition_layers=cuts, ring_depth=2,
                             compression="zfp+lz4", zfp_rate_bits=14,
                             backend="gloo", return_results=False)
        B = 1
        pipe = DistPipeline(model, cfg, (B, 64, 64, 3))
        torch.manual_seed(1234)
        inputs = [torch.randn(B, 64, 64, 3) for _ in range(steps)]
        res1, res2 = {}, {}
        feed = (lambda k: inputs[k]) if rank == 0 else None
        pipe.run(steps, feed=feed,
                 collect=lambda k, y: res1.__setitem__(k, y.clone()))
        # second run on the same rings: deferral state must reset
        pipe.run(steps, feed=feed,
                 collect=lambda k, y: res2.__setitem__(k, y.clone()))
        if rank == world - 1:
            from defer_amd.parallel.comm import Codec
            from defer_amd.parallel.partitioner import partition_model
            from defer_amd.parallel.pipeline import StageExecutor

            stages = partition_model(model, cuts)
            execs = [StageExecutor(s, "cpu", torch.float32)
                     for s in stages]

            def fwd(x):
                z = x
                for i, ex in enumerate(execs):
                    with torch.no_grad():
                        z = ex.run(z)
                    if i < len(execs) - 1:
                        c = Codec(cfg, tuple(z.shape), torch.float32,
                                  "cpu")
                        z = c.decode(c.encode(z))
                return z

            for k in range(steps):
                want = fwd(inputs[k])
                e1 = (res1[k] - want).abs().max().item()
                e2 = (res2[k] - want).abs().max().item()
                q.put(("err", k, max(e1, e2)))
            q.put(("done", rank, None))
    finally:
        dist.destroy_process_group()


def test_var_ring_deferred_interleave_gloo():
    """The deferred size/payload interleave used on RCCL, forced on gloo
    (world 3, two zfp+lz4 hops, no result return): outputs must match
    the codec-simulated chain exactly, across two sequential runs."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    steps = 4
    procs = [ctx.Process(target=_defer_worker,
                         args=(r, 3, 29721, q, steps))
             for r in range(3)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0, f"worker failed: {p.exitcode}"
    msgs = []
    while not q.empty():
        msgs.append(q.get())
    errs = [m[2] for m in msgs if m[0] == "err"]
    assert len(errs) == steps
    for e in errs:
        assert e < 1e-6, f"deferred interleave corrupted payloads: {e}"


def test_bench_self_spawn_two_ranks_cpu():
    """The driver's bare `bench.py --gpus N` form: bench self-spawns N
    ranks via mp.spawn (no torchrun) and prints one JSON line."""
    import json
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(root, "bench.py"), "--gpus", "2",
         "--device", "cpu", "--batch", "4", "--micro-batch", "2",
         "--steps", "2", "--warmup", "1", "--latency-items", "1"],
        capture_output=True, text=True, timeout=420, cwd=root)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(line) == 1, out.stdout
    d = json.loads(line[0])
    assert d["n_gpus"] == 2 and d["steps"] == 2
    assert d["config"]["parallelism"] == "pp2"
    assert d["latency_ms"]["items"] == 1


def test_bench_dual_rail_watchdog_fallback_cpu():
    """If the dual-rail warmup ever stalls (the one RCCL-only unknown),
    every rank re-execs itself single-rail on a fresh rendezvous port
    instead of hanging the driver's scaling sweep. Simulated via the
    DEFER_BENCH_TEST_STALL knob on gloo."""
    import json
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, DEFER_BENCH_TEST_STALL="1",
               DEFER_BENCH_WATCHDOG_S="2")
    out = subprocess.run(
        [sys.executable, os.path.join(root, "bench.py"), "--gpus", "3",
         "--device", "cpu", "--batch", "4", "--micro-batch", "2",
         "--steps", "2", "--warmup", "1", "--latency-items", "0",
         "--dual-rail"],
        capture_output=True, text=True, timeout=600, cwd=root, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    assert out.stderr.count("re-exec single-rail") == 3
    d = json.loads([ln for ln in out.stdout.splitlines()
                    if ln.startswith("{")][0])
    assert d["config"]["dual_rail"] is False   # fell back
    assert d["n_gpus"] == 3


def test_choose_hop_modes_unit():
    """compression="auto" rule: only hops whose raw relay exceeds the
    bottleneck stage go fp8; dual boost halves a hop's effective time."""
    from defer_amd.parallel.comm import choose_hop_modes

    # stage bottleneck 10 us; link 153 GB/s -> 1.53 MB/us... bytes in
    # B/item: hop of 2.0 MB at 153e3 B/us = 13.1 us > 10 -> fp8
    modes = choose_hop_modes([10.0, 10.0, 9.0],
                             [2.0e6, 0.4e6], 153.0)
    assert modes == ["fp8", "none"]
    # dual boost on hop 0 halves it below the bottleneck
    modes = choose_hop_modes([10.0, 10.0, 9.0],
                             [2.0e6, 0.4e6], 153.0, [2.0, 1.0])
    assert modes == ["none", "none"]


def test_dist_pipeline_auto_compression_world3():
    """compression="auto" end-to-end on gloo (world 3). With the real
    calibration the small test boundaries are far below the bottleneck
    stage, so every hop stays lossless — exercises the mode plumbing
    (all ranks must agree or sizes mismatch and the run hangs)."""
    _run(3, ["add_4", "add_12"], steps=4, compression="auto", tol=0.0)


def _auto_fp8_worker(rank, world, port, q, steps, calib_path):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)
        from defer_amd.parallel.pipeline import DistPipeline

        model = resnet50()
        cuts = ["add_4", "add_12"]
        cfg = PipelineConfig(device="cpu", dtype="fp32",
                             partition_layers=cuts, ring_depth=2,
                             compression="auto", backend="gloo",
                             calibration_file=calib_path)
        pipe = DistPipeline(model, cfg, (1, 64, 64, 3))
        assert pipe._hop_modes == ["fp8", "fp8"], pipe._hop_modes
        torch.manual_seed(1234)
        inputs = [torch.randn(1, 64, 64, 3) for _ in range(steps)]
        results = {}
        pipe.run(steps, feed=lambda k: inputs[k],
                 collect=lambda k, y: results.__setitem__(k, y.clone()))
        if rank == 0:
            # exact expectation: codec-sim the same fp8-per-hop chain
            from defer_amd.parallel.comm import Codec
            from defer_amd.parallel.partitioner import partition_model
            from defer_amd.parallel.pipeline import StageExecutor

            stages = partition_model(model, cuts)
            execs = [StageExecutor(s, "cpu", torch.float32)
                     for s in stages]

            def fwd(x):
                z = x
                for i, ex in enumerate(execs):
                    with torch.no_grad():
                        z = ex.run(z)
                    if i < len(execs) - 1:
                        c = Codec(cfg, tuple(z.shape), torch.float32,
                                  "cpu", mode="fp8")
                        z = c.decode(c.encode(z, out=c.alloc_wire()))
                return z

            for k in range(steps):
                err = (results[k] - fwd(inputs[k])).abs().max().item()
                q.put(("err", k, err))
    finally:
        dist.destroy_process_group()


def test_dist_pipeline_auto_forces_fp8_with_tiny_calibration(tmp_path):
    """A calibration that makes compute nearly free forces every hop
    past the bottleneck rule -> all-fp8 wires; outputs must match the
    fp8-sim chain near-bitwise."""
    import json

    from defer_amd.models import resnet50 as _r

    calib = {"model": "resnet50", "batch": 1, "device": "test",
             "us_per_image": {n.name: 1e-6
                              for n in _r().graph.nodes}}
    path = str(tmp_path / "tiny.json")
    with open(path, "w") as f:
        json.dump(calib, f)
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    steps = 3
    procs = [ctx.Process(target=_auto_fp8_worker,
                         args=(r, 3, 29791, q, steps, path))
             for r in range(3)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0, f"worker failed: {p.exitcode}"
    msgs = []
    while not q.empty():
        msgs.append(q.get())
    errs = [m[2] for m in msgs if m[0] == "err"]
    assert len(errs) == steps
    for e in errs:
        assert e < 1e-6, e


def test_dual_rail_with_auto_compression_world4():
    """auto wire + dual-rail combined (gloo world 4): the hop-mode
    decision must fold in the dual bandwidth boost and the split rails
    must carry whatever per-hop wire was chosen."""
    _run(4, ["add_4", "add_8", "add_12"], steps=4, compression="auto",
         dual_rail=True, tol=0.0)


def _densenet_worker(rank, q, steps):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29815"
    dist.init_process_group("gloo", rank=rank, world_size=3)
    try:
        torch.manual_seed(0)
        from defer_amd.models import densenet121
        from defer_amd.parallel.pipeline import DistPipeline

        model = densenet121(num_classes=10)
        cfg = PipelineConfig(device="cpu", dtype="fp32",
                             ring_depth=2, backend="gloo")
        pipe = DistPipeline(model, cfg, (1, 64, 64, 3))
        torch.manual_seed(77)
        xs = [torch.randn(1, 64, 64, 3) for _ in range(steps)]
        res = {}
        pipe.run(steps, feed=lambda k: xs[k],
                 collect=lambda k, y: res.__setitem__(k, y.clone()))
        if rank == 0:
            with torch.no_grad():
                for k in range(steps):
                    want = model(xs[k])
                    q.put(("err", k,
                           (res[k] - want).abs().max().item()))
    finally:
        dist.destroy_process_group()


def test_dist_pipeline_densenet_world3():
    """Concat-DAG model through the real distributed pipeline (gloo
    world 3, auto cuts inside dense blocks): partition + relay of
    concat-boundary tensors."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    steps = 2
    procs = [ctx.Process(target=_densenet_worker, args=(r, q, steps))
             for r in range(3)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0, f"worker failed: {p.exitcode}"
    msgs = []
    while not q.empty():
        msgs.append(q.get())
    errs = [m[2] for m in msgs if m[0] == "err"]
    assert len(errs) == steps
    for e in errs:
        assert e < 1e-5, e


def test_choose_hop_modes_properties():
    """Property sweep of the auto-wire rule: monotone in hop bytes,
    fp8 never chosen when every hop is under the bottleneck, boost
    only ever demotes fp8 -> none."""
    from hypothesis import given, settings, strategies as st

    from defer_amd.parallel.comm import choose_hop_modes

    @settings(max_examples=60, deadline=None)
    @given(st.lists(st.floats(0.5, 50.0), min_size=2, max_size=9),
           st.floats(10.0, 400.0))
    def check(stage_us, link):
        hop_bytes = [1e5 * (i + 1) for i in range(len(stage_us) - 1)]
        modes = choose_hop_modes(stage_us, hop_bytes, link)
        boosted = choose_hop_modes(stage_us, hop_bytes, link,
                                   [2.0] * len(hop_bytes))
        bott = max(stage_us)
        for i, m in enumerate(modes):
            expect = "fp8" if hop_bytes[i] / (link * 1e3) > bott \
                else "none"
            assert m == expect
            if m == "none":
                assert boosted[i] == "none"   # boost never adds fp8

    check()
