"""The DEFER orchestrator API (reference shape: dispatcher.py:21,107) on
CPU compute nodes — BASELINE.json config 1 plumbing."""

import queue
import threading

import torch

from defer_amd import DEFER, PipelineConfig
from defer_amd.models import resnet50


def _run_defer(model, cuts, nodes, items, cfg=None):
    eng = DEFER(nodes, config=cfg or PipelineConfig(device="cpu",
                                                    dtype="fp32"))
    in_q, out_q = queue.Queue(10), queue.Queue(10)
    t = threading.Thread(target=eng.run_defer,
                         args=(model, cuts, in_q, out_q))
    t.start()
    for x in items:
        in_q.put(x)
    in_q.put(None)
    outs = [out_q.get(timeout=120) for _ in items]
    t.join(timeout=120)
    assert not t.is_alive()
    return outs, eng


def test_defer_two_stage_cpu_matches_whole_model():
    m = resnet50()
    xs = [torch.randn(1, 64, 64, 3) for _ in range(3)]
    with torch.no_grad():
        want = [m(x) for x in xs]
    outs, eng = _run_defer(m, ["add_8"], ["cpu", "cpu"], xs)
    for o, w in zip(outs, want):
        assert torch.equal(o, w)
    assert eng.stats[0].items == 3 and eng.stats[1].items == 3


def test_defer_auto_partition_four_nodes():
    m = resnet50()
    xs = [torch.randn(1, 64, 64, 3) for _ in range(2)]
    with torch.no_grad():
        want = [m(x) for x in xs]
    outs, _ = _run_defer(m, None, ["cpu"] * 4, xs)
    for o, w in zip(outs, want):
        assert torch.equal(o, w)


def test_defer_single_node():
    m = resnet50()
    xs = [torch.randn(1, 64, 64, 3)]
    with torch.no_grad():
        want = m(xs[0])
    outs, _ = _run_defer(m, [], ["cpu"], xs)
    assert torch.equal(outs[0], want)


def test_defer_arbitrary_torch_model():
    """Capability parity with 'partition any Keras model'
    (dispatcher.py:107): an ordinary NCHW PyTorch CNN is FX-traced into
    the LayerGraph IR, auto-partitioned, and streamed through the DEFER
    orchestrator unchanged."""
    import torch.nn as nn

    class Net(nn.Module):
        def __init__(self):
            super().__init__()
            self.c1 = nn.Conv2d(3, 8, 3, padding=1)
            self.c2 = nn.Conv2d(8, 8, 3, padding=1)
            self.c3 = nn.Conv2d(8, 8, 3, padding=1)
            self.head = nn.Linear(8, 5)

        def forward(self, x):
            a = torch.relu(self.c1(x))
            b = torch.relu(self.c2(a))
            c = torch.relu(self.c3(b) + b)     # skip connection
            return self.head(c.mean(dim=(2, 3)))

    torch.manual_seed(0)
    net = Net().eval()
    xs = [torch.randn(2, 3, 16, 16) for _ in range(3)]
    with torch.no_grad():
        want = [net(x) for x in xs]
    cfg = PipelineConfig(device="cpu", dtype="fp32",
                         input_shape=(1, 3, 16, 16))
    outs, _ = _run_defer(net, None, ["cpu", "cpu"], xs, cfg=cfg)
    for o, w in zip(outs, want):
        assert torch.allclose(o, w, atol=1e-6)


def test_defer_concat_dag_model():
    """DenseNet-style torch.cat DAG through from_torch + auto-partition +
    the DEFER orchestrator: tensor inputs nested inside the cat list must
    survive FX conversion, and valid cuts must respect the long-range
    concat edges (every block consumes the original input)."""
    import torch.nn as nn

    class Dense(nn.Module):
        def __init__(self):
            super().__init__()
            self.c1 = nn.Conv2d(3, 8, 3, padding=1)
            self.c2 = nn.Conv2d(11, 8, 3, padding=1)   # cat(x, f1)
            self.c3 = nn.Conv2d(19, 8, 3, padding=1)   # cat(x, f1, f2)
            self.head = nn.Linear(8, 5)

        def forward(self, x):
            f1 = torch.relu(self.c1(x))
            f2 = torch.relu(self.c2(torch.cat([x, f1], dim=1)))
            f3 = torch.relu(self.c3(torch.cat([x, f1, f2], dim=1)))
            return self.head(f3.mean(dim=(2, 3)))

    torch.manual_seed(0)
    net = Dense().eval()
    xs = [torch.randn(2, 3, 16, 16) for _ in range(2)]
    with torch.no_grad():
        want = [net(x) for x in xs]
    cfg = PipelineConfig(device="cpu", dtype="fp32",
                         input_shape=(1, 3, 16, 16))
    outs, _ = _run_defer(net, None, ["cpu", "cpu"], xs, cfg=cfg)
    for o, w in zip(outs, want):
        assert torch.allclose(o, w, atol=1e-6)
