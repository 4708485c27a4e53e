"""defer_amd.ops — functional op API with gfx950 HIP backend.

On a GPU device every op dispatches to the hand-written CDNA4 kernels in
`defer_amd/csrc` (built in-tree as `defer_amd._hip_ops`). If a tensor is on
a CUDA device and the extension is not importable, ops raise — there is no
silent eager fallback on GPU (the HIP path is the product; the reference's
equivalent substrate was TensorFlow's native kernels, node.py:106).

On CPU, ops use the fp32 PyTorch reference implementations (tests and the
no-GPU plumbing config).
"""

from typing import Optional

import torch

from defer_amd.ops import reference as _ref

_hip = None
_hip_err: Optional[Exception] = None


def _load_hip():
    global _hip, _hip_err
    if _hip is not None or _hip_err is not None:
        return _hip
    try:
        import defer_amd._hip_ops as m  # built by build_ext / build.py

        _hip = m
    except Exception as e:  # pragma: no cover - exercised only sans build
        _hip_err = e
    return _hip


def hip_available() -> bool:
    return _load_hip() is not None


def _backend(x: torch.Tensor):
    if x.is_cuda:
        m = _load_hip()
        if m is None:
            raise RuntimeError(
                "defer_amd HIP extension (defer_amd._hip_ops) is not built "
                "but a GPU tensor was passed. Build it with "
                "`python -m defer_amd.build` (gfx950). Import error: "
                f"{_hip_err!r}"
            )
        return m
    return None


def conv2d_bn_act(x, w, scale=None, bias=None, stride=1, padding=1,
                  act="none", residual=None):
    m = _backend(x)
    if m is None:
        return _ref.conv2d_bn_act(x, w, scale, bias, stride, padding, act,
                                  residual)
    return m.conv2d_bn_act(x, w, scale, bias, residual, stride, padding,
                           act == "relu")


def conv1x1_prebn(x, w, scale, bias, out_scale=None, out_bias=None):
    """Fused relu(x*scale+bias) @ w for 1x1/s1/p0 convs (DenseNet
    BNActConv), optionally followed by the NEXT layer's folded BN-ReLU
    on the output accumulator (out_scale/out_bias): saves the
    standalone bn_act round-trips on GPU; CPU composes the reference
    ops."""
    m = _backend(x)
    if m is None:
        z = _ref.batchnorm_apply(x, scale, bias, "relu")
        y = _ref.conv2d_bn_act(z, w, None, None, 1, 0, "none", None)
        if out_scale is not None:
            y = _ref.batchnorm_apply(y, out_scale, out_bias, "relu")
        return y
    return m.conv1x1_prebn(x, w, scale, bias, out_scale, out_bias)


def batchnorm_apply(x, scale, bias, act="none"):
    m = _backend(x)
    if m is None:
        return _ref.batchnorm_apply(x, scale, bias, act)
    return m.bn_act(x, scale, bias, act == "relu")


def add_act(a, b, act="relu"):
    m = _backend(a)
    if m is None:
        return _ref.add_act(a, b, act)
    return m.add_act(a, b, act == "relu")


def relu(x):
    m = _backend(x)
    if m is None:
        return _ref.relu(x)
    return m.relu(x)


def maxpool2d(x, kernel=3, stride=2, padding=1):
    m = _backend(x)
    if m is None:
        return _ref.maxpool2d(x, kernel, stride, padding)
    return m.maxpool2d(x, kernel, stride, padding)


def avgpool2d(x, kernel=2, stride=2, padding=0):
    m = _backend(x)
    if m is None:
        return _ref.avgpool2d(x, kernel, stride, padding)
    return m.avgpool2d(x, kernel, stride, padding)


def concat_channels(xs):
    """Concat NHWC tensors on the channel (last) dim — DenseNet's dense
    connections. GPU: one strided device copy per input (no torch
    compute kernels)."""
    m = _backend(xs[0])
    if m is None:
        return _ref.concat_channels(xs)
    return m.concat_lastdim(list(xs))


def global_avg_pool(x):
    m = _backend(x)
    if m is None:
        return _ref.global_avg_pool(x)
    return m.global_avg_pool(x)


def linear(x, w, bias=None):
    m = _backend(x)
    if m is None:
        return _ref.linear(x, w, bias)
    return m.linear(x, w, bias)


def softmax(x):
    m = _backend(x)
    if m is None:
        return _ref.softmax(x)
    return m.softmax(x)
