"""GPU numerics: every gfx950 HIP kernel vs the plain PyTorch fp32
reference (defer_amd.ops.reference) on the same bf16-rounded inputs.

Shapes cover the distinct ResNet50/VGG19 op classes:
conv 1x1 s1/s2, 3x3 s1/s2, the 7x7 stem (im2col path), residual fusion,
pooling, GAP, dense, softmax.
"""

import pytest
import torch

import defer_amd.ops as ops
from defer_amd.ops import reference as ref

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _to_dev_bf16(*ts):
    return [t.to(DEV, torch.bfloat16) for t in ts]


def _relerr(got, want):
    got = got.float().cpu()
    want = want.float().cpu()
    denom = want.abs().max().clamp(min=1e-6)
    return ((got - want).abs().max() / denom).item()


requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="no GPU")


CONV_CASES = [
    # (N, H, W, Cin, Cout, R, stride, pad, act, res) — ResNet50 shape zoo
    (2, 56, 56, 64, 64, 1, 1, 0, "relu", False),     # bottleneck conv1
    (2, 56, 56, 64, 64, 3, 1, 1, "relu", False),     # bottleneck conv2
    (2, 56, 56, 64, 256, 1, 1, 0, "none", False),    # bottleneck conv3
    (2, 56, 56, 256, 512, 1, 2, 0, "none", False),   # stage proj s2
    (2, 56, 56, 128, 128, 3, 2, 1, "relu", False),   # stride-2 3x3
    (2, 14, 14, 256, 1024, 1, 1, 0, "none", True),   # fused residual
    (2, 7, 7, 512, 2048, 1, 1, 0, "none", False),    # deep 1x1 (M tail)
    (1, 224, 224, 3, 64, 7, 2, 3, "relu", False),    # stem (im2col)
    (2, 28, 28, 512, 128, 1, 1, 0, "relu", False),   # wide-in 1x1
    (1, 30, 30, 72, 40, 3, 1, 1, "none", False),     # odd Cin%8 Cout%8
]


@requires_gpu
@pytest.mark.parametrize("case", CONV_CASES,
                         ids=[f"c{i}" for i in range(len(CONV_CASES))])
def test_conv2d_bn_act(case):
    N, H, W, Cin, Cout, R, stride, pad, act, has_res = case
    x = torch.randn(N, H, W, Cin)
    w = torch.randn(Cout, R, R, Cin) * (2.0 / (Cin * R * R)) ** 0.5
    scale = torch.rand(Cout) + 0.5
    bias = torch.randn(Cout) * 0.1
    OH = (H + 2 * pad - R) // stride + 1
    res = torch.randn(N, OH, OH, Cout) if has_res else None

    xg, wg = _to_dev_bf16(x, w)
    rg = _to_dev_bf16(res)[0] if has_res else None
    want = ref.conv2d_bn_act(xg.cpu(), wg.cpu(), scale, bias, stride, pad,
                             act, rg.cpu() if has_res else None)
    got = ops.conv2d_bn_act(xg, wg, scale.to(DEV), bias.to(DEV),
                            stride=stride, padding=pad, act=act,
                            residual=rg)
    assert got.shape == want.shape
    e = _relerr(got, want)
    assert e < 0.005, f"conv rel err {e}"


def _rand_conv_cases(n=24, seed=20260914):
    """Seeded random shape sweep across the conv dispatch paths (igemm,
    gemm-mode 1x1, window, stem/channel-pad) — same generator as
    tools/tolcheck.py so measured envelopes map 1:1 to cases."""
    g = torch.Generator().manual_seed(seed)

    def ri(lo, hi):
        return int(torch.randint(lo, hi + 1, (1,), generator=g))

    cases = []
    for _ in range(n):
        R = [1, 3, 3, 7][ri(0, 3)]
        stride = ri(1, 2)
        N = ri(1, 3)
        H = ri(7, 48)
        W = ri(7, 48)
        cin_pool = [3, 8, 16, 24, 40, 64, 72, 96, 128, 192, 256]
        Cin = cin_pool[ri(0, len(cin_pool) - 1)]
        Cout = 8 * ri(1, 48)
        pad = R // 2 if ri(0, 1) else 0
        if H + 2 * pad < R or W + 2 * pad < R:
            pad = R // 2
            if H + 2 * pad < R:
                H = R
            if W + 2 * pad < R:
                W = R
        act = "relu" if ri(0, 1) else "none"
        has_res = bool(ri(0, 1)) and R == 1 and stride == 1 and pad == 0
        cases.append((N, H, W, Cin, Cout, R, stride, pad, act, has_res))
    return cases


@requires_gpu
@pytest.mark.parametrize("case", _rand_conv_cases(),
                         ids=[f"s{i}" for i in range(24)])
def test_conv_shape_sweep(case):
    """Property-style shape sweep: all dispatch paths, tolerance set
    from the measured envelope (tools/tolcheck.py), ~5x headroom."""
    N, H, W, Cin, Cout, R, stride, pad, act, has_res = case
    torch.manual_seed(hash((N, H, W, Cin, Cout, R, stride)) % 2**31)
    x = torch.randn(N, H, W, Cin)
    w = torch.randn(Cout, R, R, Cin) * (2.0 / (Cin * R * R)) ** 0.5
    scale = torch.rand(Cout) + 0.5
    bias = torch.randn(Cout) * 0.1
    OH = (H + 2 * pad - R) // stride + 1
    OW = (W + 2 * pad - R) // stride + 1
    res = torch.randn(N, OH, OW, Cout) if has_res else None
    xg, wg = _to_dev_bf16(x, w)
    rg = _to_dev_bf16(res)[0] if has_res else None
    want = ref.conv2d_bn_act(xg.cpu(), wg.cpu(), scale, bias, stride,
                             pad, act, rg.cpu() if has_res else None)
    got = ops.conv2d_bn_act(xg, wg, scale.to(DEV), bias.to(DEV),
                            stride=stride, padding=pad, act=act,
                            residual=rg)
    assert got.shape == want.shape
    e = _relerr(got, want)
    assert e < 0.005, f"conv sweep rel err {e} for {case}"


@requires_gpu
def test_conv_zero_padding_boundary():
    # all-ones input, 3x3: border sums differ from interior — catches
    # wrong pad predication
    x = torch.ones(1, 8, 8, 64)
    w = torch.ones(64, 3, 3, 64) * 0.01
    xg, wg = _to_dev_bf16(x, w)
    want = ref.conv2d_bn_act(xg.cpu(), wg.cpu(), None, None, 1, 1, "none")
    got = ops.conv2d_bn_act(xg, wg, None, None, stride=1, padding=1)
    assert _relerr(got, want) < 0.005


@requires_gpu
def test_maxpool():
    x = torch.randn(2, 112, 112, 64)
    (xg,) = _to_dev_bf16(x)
    want = ref.maxpool2d(xg.cpu(), 3, 2, 1)
    got = ops.maxpool2d(xg, 3, 2, 1)
    assert torch.equal(got.cpu(), want)  # max of bf16 values is exact


@requires_gpu
def test_maxpool_vgg():
    x = torch.randn(2, 56, 56, 128)
    (xg,) = _to_dev_bf16(x)
    want = ref.maxpool2d(xg.cpu(), 2, 2, 0)
    got = ops.maxpool2d(xg, 2, 2, 0)
    assert torch.equal(got.cpu(), want)


@requires_gpu
def test_global_avg_pool():
    x = torch.randn(3, 7, 7, 2048)
    (xg,) = _to_dev_bf16(x)
    want = ref.global_avg_pool(xg.cpu())
    got = ops.global_avg_pool(xg)
    assert _relerr(got, want) < 0.01


@requires_gpu
def test_linear():
    x = torch.randn(64, 2048)
    w = torch.randn(1000, 2048) * (1 / 2048) ** 0.5
    b = torch.randn(1000)
    xg, wg = _to_dev_bf16(x, w)
    want = ref.linear(xg.cpu(), wg.cpu(), b)
    got = ops.linear(xg, wg, b.to(DEV))
    assert _relerr(got, want) < 0.005


@requires_gpu
def test_linear_large_k_vgg_fc1():
    """The faithful VGG19 fc1 (25088x4096): large-K accumulation path of
    the MFMA GEMM."""
    x = torch.randn(8, 25088)
    w = torch.randn(4096, 25088) * (1 / 25088) ** 0.5
    b = torch.randn(4096)
    xg, wg = _to_dev_bf16(x, w)
    want = ref.linear(xg.cpu(), wg.cpu(), b)
    got = ops.linear(xg, wg, b.to(DEV))
    assert _relerr(got, want) < 0.005


@requires_gpu
def test_vgg19_full_forward_vs_cpu():
    """Whole faithful-head VGG19 bf16 GPU forward vs fp32 CPU reference
    (flatten -> 25088x4096 fc1 in the real graph)."""
    from defer_amd.graph import GraphModel
    from defer_amd.models import vgg19
    from defer_amd.parallel.pipeline import StageExecutor

    torch.manual_seed(0)
    m = vgg19()
    x = torch.randn(2, 224, 224, 3)
    with torch.no_grad():
        want = m(x).float()
    ex = StageExecutor(GraphModel(m.graph), DEV, torch.bfloat16,
                       use_graph=False)
    with torch.no_grad():
        got = ex.run(x.to(DEV, torch.bfloat16)).float().cpu()
    cos = torch.nn.functional.cosine_similarity(got, want, dim=-1)
    assert (cos > 0.98).all(), f"cosine {cos}"


@requires_gpu
def test_softmax():
    x = torch.randn(64, 1000) * 4
    (xg,) = _to_dev_bf16(x)
    want = ref.softmax(xg.cpu())
    got = ops.softmax(xg)
    assert (got.float().cpu() - want.float()).abs().max() < 2e-3
    assert ((got.float().sum(-1) - 1).abs() < 2e-2).all()


@requires_gpu
def test_add_act_bn_act_relu():
    a = torch.randn(2, 14, 14, 1024)
    b = torch.randn(2, 14, 14, 1024)
    ag, bg = _to_dev_bf16(a, b)
    assert _relerr(ops.add_act(ag, bg, "relu"),
                   ref.add_act(ag.cpu(), bg.cpu(), "relu")) < 0.01
    sc = torch.rand(1024) + 0.5
    bi = torch.randn(1024)
    assert _relerr(ops.batchnorm_apply(ag, sc.to(DEV), bi.to(DEV), "relu"),
                   ref.batchnorm_apply(ag.cpu(), sc, bi, "relu")) < 0.01
    assert torch.equal(ops.relu(ag).cpu(), ref.relu(ag.cpu()))


@requires_gpu
def test_resnet50_full_forward_vs_cpu():
    """Whole-model bf16 GPU forward vs fp32 CPU reference: logits must
    correlate and the top-1 class agree for most inputs."""
    from defer_amd.graph import GraphModel
    from defer_amd.models import resnet50
    from defer_amd.parallel.pipeline import StageExecutor

    torch.manual_seed(0)
    m = resnet50()
    x = torch.randn(4, 224, 224, 3)
    with torch.no_grad():
        want = m(x).float()
    ex = StageExecutor(GraphModel(m.graph), DEV, torch.bfloat16,
                       use_graph=False)
    with torch.no_grad():
        got = ex.run(x.to(DEV, torch.bfloat16)).float().cpu()
    cos = torch.nn.functional.cosine_similarity(got, want, dim=-1)
    assert (cos > 0.98).all(), f"cosine {cos}"
    top_match = (got.argmax(-1) == want.argmax(-1)).float().mean()
    assert top_match >= 0.75, f"top1 agreement {top_match}"


@requires_gpu
def test_defer_threaded_multistage_on_gpu():
    """4-stage DEFER pipeline chained on one GPU (stage handoff via
    device tensors) matches the single-stage GPU forward."""
    import queue
    import threading

    from defer_amd import DEFER, PipelineConfig
    from defer_amd.graph import GraphModel
    from defer_amd.models import resnet50
    from defer_amd.parallel.pipeline import StageExecutor

    torch.manual_seed(0)
    m = resnet50()
    ex = StageExecutor(GraphModel(m.graph), DEV, torch.bfloat16)
    xs = [torch.randn(2, 224, 224, 3) for _ in range(3)]
    with torch.no_grad():
        want = [ex.run(x.to(DEV, torch.bfloat16)).float().cpu()
                for x in xs]

    eng = DEFER(["cuda:0"] * 4,
                config=PipelineConfig(device="cuda", dtype="bf16"))
    in_q, out_q = queue.Queue(8), queue.Queue(8)
    t = threading.Thread(target=eng.run_defer,
                         args=(resnet50_copy_like(m), None, in_q, out_q))
    t.start()
    for x in xs:
        in_q.put(x.to(DEV, torch.bfloat16))
    in_q.put(None)
    outs = [out_q.get(timeout=180) for _ in xs]
    t.join(timeout=60)
    for o, w in zip(outs, want):
        assert o.shape == w.shape
        assert (o - w).abs().max() < 1e-2


def resnet50_copy_like(m):
    """Fresh resnet50 with the same weights (same seed construction)."""
    import torch as _t

    from defer_amd.models import resnet50 as _r

    _t.manual_seed(0)
    return _r()


@pytest.mark.gpu
def test_stage_stats_event_timer_gpu():
    """hipEvent per-stage timing (utils/trace.py) measures real device
    time on a single-stage DistPipeline run."""
    import os

    import torch.distributed as dist

    from defer_amd.config import PipelineConfig
    from defer_amd.models import resnet50
    from defer_amd.parallel.pipeline import DistPipeline

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29784")
        dist.init_process_group("gloo", rank=0, world_size=1)
    cfg = PipelineConfig(device="cuda", dtype="bf16", batch_size=8,
                         use_hip_graphs=False, backend="gloo",
                         log_stage_stats=True)
    pipe = DistPipeline(resnet50(), cfg, (8, 224, 224, 3),
                        device=torch.device("cuda", 0))
    pipe.run(3, feed=lambda k: torch.randn(
        8, 224, 224, 3, device="cuda", dtype=torch.bfloat16),
        collect=lambda k, y: None)
    torch.cuda.synchronize()
    assert pipe.stats.items == 3
    ms = pipe.stats.compute_ms
    assert 0.05 < ms < 5000, ms   # real device time, not zero


@pytest.mark.gpu
def test_window_conv_forced_numerics():
    """The window-reuse 3x3 path vs the fp32 reference, FORCED past its
    size gate (DEFER_CONV_VARIANT=w is latched at first kernel launch,
    so this runs in a subprocess)."""
    import os
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, DEFER_CONV_VARIANT="w")
    out = subprocess.run(
        [sys.executable, os.path.join(root, "tools", "wincheck.py")],
        capture_output=True, text=True, timeout=600, env=env, cwd=root)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-1000:]


@pytest.mark.gpu
def test_stage_executor_hipgraph_capture_single_thread():
    """The hipGraph capture path (bench --graphs; safe single-threaded)
    matches eager execution."""
    from defer_amd.graph import GraphModel
    from defer_amd.models import resnet50
    from defer_amd.parallel.pipeline import StageExecutor

    torch.manual_seed(0)
    m = resnet50()
    eager = StageExecutor(GraphModel(m.graph), DEV, torch.bfloat16,
                          use_graph=False)
    torch.manual_seed(0)
    graphed = StageExecutor(GraphModel(resnet50().graph), DEV,
                            torch.bfloat16, use_graph=True)
    x = torch.randn(2, 224, 224, 3, device=DEV, dtype=torch.bfloat16)
    with torch.no_grad():
        want = eager.run(x).float().cpu()
        _ = graphed.run(x)            # first call captures (or falls back)
        got = graphed.run(x).float().cpu()
    torch.cuda.synchronize()
    assert torch.allclose(got, want, atol=1e-3), \
        float((got - want).abs().max())


@requires_gpu
def test_avgpool_gpu():
    """DenseNet transition 2x2/2 average pool vs the fp32 reference
    (bf16 mean of 4 values: tight tolerance)."""
    for shape, k, s, p in [((2, 56, 56, 128), 2, 2, 0),
                           ((3, 14, 14, 512), 2, 2, 0),
                           ((1, 9, 9, 64), 3, 2, 1)]:
        x = torch.randn(*shape)
        (xg,) = _to_dev_bf16(x)
        want = ref.avgpool2d(xg.cpu(), k, s, p)
        got = ops.avgpool2d(xg, k, s, p)
        assert got.shape == want.shape
        assert _relerr(got, want) < 0.005


@requires_gpu
def test_concat_channels_gpu():
    """Channel concat via strided device copies: bitwise."""
    a = torch.randn(2, 7, 7, 8).to(DEV, torch.bfloat16)
    b = torch.randn(2, 7, 7, 24).to(DEV, torch.bfloat16)
    c = torch.randn(2, 7, 7, 64).to(DEV, torch.bfloat16)
    got = ops.concat_channels([a, b, c])
    want = torch.cat([a, b, c], dim=-1)
    assert torch.equal(got.cpu(), want.cpu())


@requires_gpu
def test_densenet121_full_forward_vs_cpu():
    """Whole DenseNet-121 bf16 GPU forward (BN-ReLU-Conv composites,
    concat relay, avg-pool transitions) vs fp32 CPU reference."""
    from defer_amd.graph import GraphModel
    from defer_amd.models import densenet121
    from defer_amd.parallel.pipeline import StageExecutor

    torch.manual_seed(0)
    m = densenet121()
    x = torch.randn(2, 224, 224, 3)
    with torch.no_grad():
        want = m(x).float()
    ex = StageExecutor(GraphModel(m.graph), DEV, torch.bfloat16,
                       use_graph=False)
    with torch.no_grad():
        got = ex.run(x.to(DEV, torch.bfloat16)).float().cpu()
    cos = torch.nn.functional.cosine_similarity(got, want, dim=-1)
    assert (cos > 0.98).all(), f"cosine {cos}"


@requires_gpu
def test_conv1x1_prebn_bitwise_vs_two_step():
    """The fused pre-activation 1x1 (gemm_prebn_kernel) must be
    BIT-identical to bn_act -> conv2d_bn_act: same fp32 bn + bf16
    rounding before the MFMA, same ascending-k 16x16x32 accumulation."""
    torch.manual_seed(4)
    for N, H, W, Cin, Cout in [(2, 28, 28, 96, 128),    # K % 64 != 0
                               (2, 14, 14, 256, 128),
                               (1, 7, 7, 1024, 512),
                               (3, 8, 8, 64, 40)]:      # Cout % 64 != 0
        x = torch.randn(N, H, W, Cin).to(DEV, torch.bfloat16)
        w = (torch.randn(Cout, 1, 1, Cin) * 0.05).to(DEV, torch.bfloat16)
        sc = (torch.rand(Cin) + 0.5).to(DEV)
        bi = (torch.randn(Cin) * 0.1).to(DEV)
        z = ops.batchnorm_apply(x, sc, bi, act="relu")
        want = ops.conv2d_bn_act(z, w, None, None, stride=1, padding=0,
                                 act="none")
        got = ops.conv1x1_prebn(x, w, sc, bi)
        assert got.shape == want.shape
        assert torch.equal(got, want), (
            N, H, W, Cin, Cout,
            (got.float() - want.float()).abs().max().item())


@requires_gpu
def test_conv1x1_prebn_out_affine():
    """The output-affine variant (next layer's folded BN-ReLU on the
    fp32 accumulator): compare against the two-step composition with a
    tolerance (the fused form skips the intermediate bf16 rounding, so
    it is slightly MORE accurate, not bit-equal)."""
    torch.manual_seed(6)
    N, H, W, Cin, Cout = 2, 14, 14, 160, 128
    x = torch.randn(N, H, W, Cin).to(DEV, torch.bfloat16)
    w = (torch.randn(Cout, 1, 1, Cin) * 0.05).to(DEV, torch.bfloat16)
    sc = (torch.rand(Cin) + 0.5).to(DEV)
    bi = (torch.randn(Cin) * 0.1).to(DEV)
    osc = (torch.rand(Cout) + 0.5).to(DEV)
    obi = (torch.randn(Cout) * 0.1).to(DEV)
    y1 = ops.conv1x1_prebn(x, w, sc, bi)
    want = ops.batchnorm_apply(y1, osc, obi, act="relu")
    got = ops.conv1x1_prebn(x, w, sc, bi, osc, obi)
    assert got.shape == want.shape
    # the two-step `want` carries one extra bf16 rounding of the
    # intermediate (~0.4% per value); 0.008 covers exactly that
    assert _relerr(got, want) < 0.008
