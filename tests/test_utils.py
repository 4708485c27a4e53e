"""Aux-subsystem tests: partition visualization (plot_model parity,
reference node.py:39), per-stage event tracing, fail-fast stage errors
(SURVEY.md §5)."""

import queue
import threading

import torch

from defer_amd import DEFER, PipelineConfig
from defer_amd.models import resnet50
from defer_amd.parallel.partitioner import as_graph_model, partition_model
from defer_amd.utils.trace import EventTimer
from defer_amd.utils.visualize import describe, dump_partition, to_dot


def test_dot_and_describe():
    gm = as_graph_model(resnet50())
    dot = to_dot(gm.graph, name="resnet50", cut_points=["add_8"])
    assert dot.startswith('digraph "resnet50"')
    assert '"add_8"' in dot and "peripheries=2" in dot
    # every node appears, every edge references defined nodes
    for n in gm.graph.nodes:
        assert f'"{n.name}"' in dot
    txt = describe(gm.graph, name="resnet50")
    assert "layers" in txt and "params" in txt
    assert "add_8" in txt


def test_dump_partition(tmp_path):
    gm = as_graph_model(resnet50())
    stages = partition_model(gm, ["add_4", "add_8"])
    paths = dump_partition(stages, str(tmp_path), ["add_4", "add_8"])
    assert len(paths) == 2 * 3 + 1
    overview = (tmp_path / "partition.txt").read_text()
    assert "stage 0" in overview and "stage 2" in overview
    assert "add_4" in overview
    # stage dumps are valid-looking DOT
    assert (tmp_path / "stage_1.dot").read_text().startswith("digraph")


def test_event_timer_cpu():
    import time

    t = EventTimer(torch.device("cpu"))
    for _ in range(3):
        t.start()
        time.sleep(0.01)
        t.stop()
    assert t.count == 3
    assert 20 < t.total_ms() < 500
    assert t.mean_ms() > 5


def test_defer_stage_failure_fails_fast():
    class Boom(torch.nn.Module):
        def forward(self, x):
            raise ValueError("kaboom")

    from defer_amd.graph import GraphModel, GraphNode, LayerGraph

    g = LayerGraph([GraphNode("ok", torch.nn.Identity(), ["input"]),
                    GraphNode("boom", Boom(), ["ok"])])
    eng = DEFER(["cpu", "cpu"],
                config=PipelineConfig(device="cpu", dtype="fp32"))
    in_q, out_q = queue.Queue(4), queue.Queue(4)
    err = []

    def run():
        try:
            eng.run_defer(GraphModel(g), ["ok"], in_q, out_q)
        except RuntimeError as e:
            err.append(e)

    t = threading.Thread(target=run)
    t.start()
    for _ in range(3):
        in_q.put(torch.randn(2, 4))
    in_q.put(None)
    t.join(timeout=60)
    assert not t.is_alive(), "failed stage hung the chain"
    assert err and "stage 1" in str(err[0]) and "kaboom" in str(err[0])


def test_dist_pipeline_partition_dump(tmp_path):
    # world=1 path exercises the dump hook without a process group
    import torch.distributed as dist

    if not dist.is_initialized():
        import os

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29781")
        dist.init_process_group("gloo", rank=0, world_size=1)
    from defer_amd.parallel.pipeline import DistPipeline

    cfg = PipelineConfig(device="cpu", dtype="fp32", batch_size=2,
                         use_hip_graphs=False, backend="gloo",
                         log_stage_stats=True,
                         partition_dump_dir=str(tmp_path / "dump"))
    pipe = DistPipeline(resnet50(), cfg, (2, 64, 64, 3),
                        device=torch.device("cpu"))
    pipe.run(2, feed=lambda k: torch.randn(2, 64, 64, 3),
             collect=lambda k, y: None)
    assert (tmp_path / "dump" / "partition.txt").exists()
    assert pipe.stats.compute_ms > 0
    pipe.reset_stats()
    assert pipe.stats.compute_ms == 0


def test_scale_report_tool(tmp_path):
    import json
    import os
    import subprocess
    import sys

    rows = [dict(metric="images/sec (whole node) ResNet50 pipeline",
                 value=v, unit="images/sec", n_gpus=n, ms_per_step=1.0,
                 dtype="bf16", data="synthetic",
                 config=dict(model="resnet50", global_batch=64,
                             parallelism=f"pp{n}"))
            for n, v in [(1, 1000.0), (2, 1900.0), (4, 3500.0)]]
    p = tmp_path / "runs.jsonl"
    p.write_text("".join(json.dumps(r) + "\n" for r in rows))
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(root, "tools", "scale_report.py"),
         str(p)], capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    assert "1.90x" in out.stdout and "95.0%" in out.stdout
    assert "+250%" in out.stdout


def test_visualize_function_layers():
    """DOT/describe handle non-module (lambda / function) layers — the
    FX frontend produces them for `x + y`, relu, etc."""
    from defer_amd.graph import GraphNode, LayerGraph

    g = LayerGraph([
        GraphNode("lin", torch.nn.Linear(4, 4), ["input"]),
        GraphNode("add", lambda a, b: a + b, ["lin", "input"]),
    ])
    dot = to_dot(g, name="fn")
    assert '"add"' in dot and "digraph" in dot
    txt = describe(g, name="fn")
    assert "add" in txt and "<lambda>" in txt or "fn" in txt


def test_powerbench_tool_wraps_json_line(tmp_path):
    """tools/powerbench.py relays the wrapped command's JSON line and
    augments it with power fields (null watts where rocm-smi has no GPU),
    and propagates a non-zero exit code."""
    import json
    import os
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    tool = os.path.join(root, "tools", "powerbench.py")
    inner = ("import json; print('banner noise'); "
             "print(json.dumps({'metric': 'm', 'value': 10.0, "
             "'unit': 'images/sec'}))")
    out = subprocess.run(
        [sys.executable, tool, "--", sys.executable, "-c", inner],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    row = json.loads(out.stdout.strip().splitlines()[-1])
    assert row["metric"] == "m" and row["value"] == 10.0
    assert "avg_watts" in row and "power_samples" in row

    fail = subprocess.run(
        [sys.executable, tool, "--", sys.executable, "-c",
         "import sys; sys.exit(3)"],
        capture_output=True, text=True, timeout=120)
    assert fail.returncode == 3


def test_powerbench_parses_rocm_smi_json():
    """read_power_w sums per-card socket power from rocm-smi --json
    output (skipping caps/max fields)."""
    import importlib.util
    import json
    import os
    import sys
    from unittest import mock

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    spec = importlib.util.spec_from_file_location(
        "powerbench", os.path.join(root, "tools", "powerbench.py"))
    pb = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(pb)

    fake = json.dumps({
        "card0": {"Average Graphics Package Power (W)": "312.0",
                  "Max Graphics Package Power (W)": "1400.0"},
        "card1": {"Current Socket Graphics Package Power (W)": "88.5"},
    })
    done = mock.Mock()
    done.stdout = fake
    with mock.patch.object(pb.subprocess, "run", return_value=done):
        assert abs(pb.read_power_w() - 400.5) < 1e-9


def test_dual_rail_link_assignment_is_collision_free():
    """comm.hop_via's claim, checked exhaustively for world 3..16: across
    all hops' rails (direct link + the two forwarder links) plus the
    result-return link, every DIRECTED xGMI link is used at most once —
    each rail therefore runs at full link rate."""
    from defer_amd.parallel.comm import hop_via

    for world in range(3, 17):
        links = []
        for i in range(world - 1):
            links.append((i, i + 1))          # direct rail
            via = hop_via(i, world)
            if via is not None:
                assert via not in (i, i + 1)
                links.append((i, via))        # src -> forwarder
                links.append((via, i + 1))    # forwarder -> dst
        links.append((world - 1, 0))          # result return
        assert len(links) == len(set(links)), (world, links)
        # every hop is dual-rail except the last one at world == 3
        n_single = sum(1 for i in range(world - 1)
                       if hop_via(i, world) is None)
        assert n_single == (1 if world == 3 else 0)


def test_dual_rail_split_point():
    from defer_amd.parallel.comm import split_point

    assert split_point(2) == 1
    assert split_point(100) == 50
    assert split_point(1 << 20) == (1 << 19)
    n = 1000003
    s = split_point(n)
    assert 0 < s < n and s % 64 == 0


def test_http_serving_example():
    """examples/serve_http.py: REST front-end over the DEFER pipeline
    (2 CPU stages) answers concurrent requests with the same outputs as
    the bare model."""
    import importlib.util
    import os
    import sys
    from concurrent.futures import ThreadPoolExecutor

    from fastapi.testclient import TestClient

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    spec = importlib.util.spec_from_file_location(
        "serve_http", os.path.join(root, "examples", "serve_http.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)

    torch.manual_seed(0)
    app = mod.build_app(["cpu", "cpu"], input_hw=64)
    from defer_amd.models import resnet50
    torch.manual_seed(0)
    ref = resnet50()

    client = TestClient(app)
    torch.manual_seed(11)
    xs = [torch.randn(1, 64, 64, 3) for _ in range(4)]

    def post(x):
        r = client.post("/infer", json={"data": x.flatten().tolist(),
                                        "shape": list(x.shape)})
        assert r.status_code == 200, r.text
        return torch.tensor(r.json()["probs"])

    with ThreadPoolExecutor(4) as pool:
        outs = list(pool.map(post, xs))
    with torch.no_grad():
        for x, y in zip(xs, outs):
            want = ref(x)
            assert torch.allclose(y, want, atol=1e-4), \
                (y - want).abs().max()

    bad = client.post("/infer", json={"data": [1.0], "shape": [1]})
    assert bad.status_code == 400


def test_stream_infer_example_runs():
    """examples/stream_infer.py (the reference's test/test.py usage
    shape) runs end to end on CPU."""
    import os
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(root, "examples",
                                      "stream_infer.py"),
         "--device", "cpu", "--items", "3", "--batch", "1"],
        capture_output=True, text=True, timeout=300, cwd=root)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "images/sec" in out.stdout


def test_pipeline_config_validation():
    import pytest

    from defer_amd.config import PipelineConfig

    PipelineConfig()                      # defaults valid
    with pytest.raises(ValueError, match="compression"):
        PipelineConfig(compression="gzip")
    with pytest.raises(ValueError, match="dtype"):
        PipelineConfig(dtype="fp64")
    with pytest.raises(ValueError, match="ring_depth"):
        PipelineConfig(ring_depth=0)
    with pytest.raises(ValueError, match="zfp_rate_bits"):
        PipelineConfig(zfp_rate_bits=40)
