// Torch bindings for the defer_amd gfx950 kernel library.
//
// Host-only TU: checks device/dtype/contiguity and calls the launchers in
// the .hip TUs on the current HIP stream, so ops compose with torch.cuda
// streams and hipGraph capture.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <functional>
#include <mutex>
#include <unordered_map>

#include "kernels.h"

namespace {

using at::Tensor;
using defer_hip::ConvParams;

hipStream_t cur_stream() {
    return at::cuda::getCurrentHIPStream().stream();
}

const void* bptr(const Tensor& t) { return t.data_ptr(); }
void* bptr_mut(Tensor& t) { return t.data_ptr(); }

void check_bf16(const Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
    TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
    TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// 16B zero source for OOB global_load_lds lanes, one per process
const void* zero_buf() {
    static Tensor z = at::zeros(
        {16}, at::TensorOptions().dtype(at::kBFloat16).device(at::kCUDA));
    return bptr(z);
}

// scale/bias must be non-null in the kernel (nullable per-element loads
// de-pipeline the epilogue); substitute cached ones/zeros when absent.
const float* ones_buf() {
    static Tensor t = at::ones(
        {8192}, at::TensorOptions().dtype(at::kFloat).device(at::kCUDA));
    return t.data_ptr<float>();
}
const float* zeros_buf() {
    static Tensor t = at::zeros(
        {8192}, at::TensorOptions().dtype(at::kFloat).device(at::kCUDA));
    return t.data_ptr<float>();
}

// Prepared-weight cache for the stem paths (K-order repack / channel
// pad). Weights are constant in inference, so the one-time transform is
// keyed on (data_ptr, numel, variant) — the reference pays its weight
// prep once at dispatch too (node.py:34). The SOURCE tensor is pinned
// in the entry: the caching allocator would otherwise reuse a freed
// weight's pointer for a different model's same-shaped weights and the
// key would collide with stale data (observed as a flaky cross-test
// mismatch). Unbounded but tiny (one entry per stem conv weight);
// in-place weight mutation is not detected (inference engine — the
// reference also ships weights exactly once, dispatcher.py:57).
struct PrepEntry {
    Tensor src;       // pinning src keeps its pointer unique
    Tensor prepped;
    uint32_t version; // src._version at prep time: in-place mutation of
                      // the source weight (e.g. checkpoint reload into
                      // the same tensors) bumps it and invalidates the
                      // cached repack
};

Tensor cached_weight_prep(const Tensor& w, int64_t variant,
                          const std::function<Tensor()>& make) {
    static std::unordered_map<uint64_t, PrepEntry> cache;
    static std::mutex mu;
    uint64_t key = (uint64_t)(uintptr_t)w.data_ptr() * 31 +
                   (uint64_t)w.numel() * 7 + (uint64_t)variant;
    uint32_t ver = w.unsafeGetTensorImpl()->version_counter().current_version();
    std::lock_guard<std::mutex> g(mu);
    auto it = cache.find(key);
    if (it != cache.end() &&
        it->second.src.data_ptr() == w.data_ptr() &&
        it->second.src.numel() == w.numel() &&
        it->second.version == ver)
        return it->second.prepped;
    Tensor t = make();
    cache[key] = {w, t, ver};
    return t;
}

const float* fptr_opt(const c10::optional<Tensor>& t, const char* name,
                      const float* fallback, int64_t n) {
    if (!t) {
        TORCH_CHECK(n <= 8192, "Cout too large for cached scale/bias");
        return fallback;
    }
    TORCH_CHECK(t->scalar_type() == at::kFloat, name, " must be fp32");
    TORCH_CHECK(t->is_contiguous(), name, " must be contiguous");
    return t->data_ptr<float>();
}

Tensor conv2d_bn_act(Tensor x, Tensor w, c10::optional<Tensor> scale,
                     c10::optional<Tensor> bias, c10::optional<Tensor> res,
                     int64_t stride, int64_t pad, bool relu) {
    check_bf16(x, "x");
    check_bf16(w, "w");
    TORCH_CHECK(x.dim() == 4, "x must be NHWC");
    TORCH_CHECK(w.dim() == 4, "w must be OHWI");
    int NB = x.size(0), H = x.size(1), W = x.size(2), Cin = x.size(3);
    int Cout = w.size(0), R = w.size(1), S = w.size(2);
    TORCH_CHECK(w.size(3) == Cin, "w Cin mismatch");
    int OH = (int)((H + 2 * pad - R) / stride + 1);
    int OW = (int)((W + 2 * pad - S) / stride + 1);
    long M = (long)NB * OH * OW;
    TORCH_CHECK(M < (1LL << 31), "M too large");
    auto out = at::empty({NB, OH, OW, Cout}, x.options());
    if (res) check_bf16(*res, "res");
    hipStream_t s = cur_stream();

    ConvParams p{};
    p.scale = fptr_opt(scale, "scale", ones_buf(), Cout);
    p.bias = fptr_opt(bias, "bias", zeros_buf(), Cout);
    p.res = res ? bptr(*res) : nullptr;
    p.out = bptr_mut(out);
    p.zbuf = zero_buf();
    p.M = (int)M;
    p.Cout = Cout;

    if (Cin % 8 != 0) {
        // small-Cin window path (VGG-style 3x3 stems): channel-pad to
        // 8, repack weights j-major (cached), stage each 8x16 output
        // tile's input window to LDS once — kills the ~9x input
        // re-reads of the channel-pad gather path (b1.c1 143 -> 91 us).
        // The 7x7/s2 ResNet stem stays on the spatial-prepad run-packed
        // path: its K is 168 vs the window kernel's zero-padded 448
        // (Cin 3 -> 8), and the 2.7x extra MFMA work measured slower.
        if (!res && R == S && R == 3 &&
            (stride == 1 || stride == 2) && Cout == 64) {
            int Kpad = (R * R * 8 + 63) / 64 * 64;
            auto xp = at::empty({NB, H, W, 8}, x.options());
            defer_hip::launch_pad_channels(bptr(x), bptr_mut(xp),
                                           (long)NB * H * W, Cin, 8, s);
            auto wp = cached_weight_prep(w, 1000 + Kpad, [&] {
                auto t = at::empty({Cout, Kpad}, w.options());
                defer_hip::launch_swin_repack_w(bptr(w), bptr_mut(t),
                                               Cout, R, Cin, Kpad, s);
                return t;
            });
            ConvParams q = p;
            q.x = bptr(xp);
            q.w = bptr(wp);
            q.NB = NB; q.H = H; q.W = W; q.Cin = 8;
            q.OH = OH; q.OW = OW; q.R = R; q.S = S;
            q.stride = (int)stride; q.pad = (int)pad;
            q.K = R * S * 8;
            if (defer_hip::launch_conv_swin(q, relu, R, (int)stride, s))
                return out;
        }
        if ((stride * Cin) % 2 == 0) {
            // STEM path (ResNet 7x7/s2, Cin=3): spatially pre-pad the
            // raw input and gather K as (r, run)-major where a run is
            // TR = ceil(S*Cin/8)*8 CONTIGUOUS bf16 of one padded input
            // row (S*Cin real + a few next-pixel values that multiply
            // zero weights). K drops from R*S*pad8(Cin) (392) to R*TR
            // (168) and the gather needs no validity math at all.
            int TR = (S * Cin + 7) / 8 * 8;
            int Keff = R * TR;
            int nk = (Keff + 63) / 64;
            int rmax = (nk * 64 - 1) / TR;          // last k row touched
            int PW = std::max(W + 2 * (int)pad,
                              (OW - 1) * (int)stride
                                  + (TR + Cin - 1) / Cin);
            int PH = std::max(H + 2 * (int)pad,
                              (OH - 1) * (int)stride + rmax + 1);
            if (((long)PW * Cin) % 2) PW += 1;      // 4-B chunk alignment
            auto xp = at::empty({NB, PH, PW, Cin}, x.options());
            defer_hip::launch_pad2d(bptr(x), bptr_mut(xp), NB, H, W, Cin,
                                    PH, PW, (int)pad, (int)pad, s);
            auto wp = cached_weight_prep(w, TR, [&] {
                auto t = at::empty({Cout, Keff}, w.options());
                defer_hip::launch_stem_repack_w(bptr(w), bptr_mut(t),
                                                Cout, R, S, Cin, TR, s);
                return t;
            });
            p.x = bptr(xp);
            p.w = bptr(wp);
            p.K = Keff;
            p.NB = NB; p.H = PH; p.W = PW; p.Cin = Cin;
            p.OH = OH; p.OW = OW; p.R = R;
            p.S = TR;                   // S carries the run length TR
            p.stride = (int)stride; p.pad = 0;
            defer_hip::launch_conv_igemm(p, relu, (bool)res, false, true,
                                         s);
            return out;
        }
        // odd stride*Cin (e.g. 3x3/s1 VGG stem): zero-pad channels to 8
        // and take the regular implicit-GEMM path.
        int C8 = (Cin + 7) / 8 * 8;
        auto xp = at::empty({NB, H, W, C8}, x.options());
        defer_hip::launch_pad_channels(bptr(x), bptr_mut(xp),
                                       (long)NB * H * W, Cin, C8, s);
        auto wp = cached_weight_prep(w, -C8, [&] {
            auto t = at::empty({Cout, R, S, C8}, w.options());
            defer_hip::launch_pad_channels(bptr(w), bptr_mut(t),
                                           (long)Cout * R * S, Cin, C8, s);
            return t;
        });
        p.x = bptr(xp);
        p.w = bptr(wp);
        p.K = R * S * C8;
        p.NB = NB; p.H = H; p.W = W; p.Cin = C8;
        p.OH = OH; p.OW = OW; p.R = R; p.S = S;
        p.stride = (int)stride; p.pad = (int)pad;
        defer_hip::launch_conv_igemm(p, relu, (bool)res, false, false, s);
        return out;
    }

    p.x = bptr(x);
    p.w = bptr(w);
    p.K = R * S * Cin;
    p.NB = NB; p.H = H; p.W = W; p.Cin = Cin;
    p.OH = OH; p.OW = OW; p.R = R; p.S = S;
    p.stride = (int)stride; p.pad = (int)pad;
    bool gemm_mode = (R == 1 && S == 1 && stride == 1 && pad == 0);
    if (gemm_mode) {
        p.NB = (int)M; p.H = 1; p.W = 1; p.Cin = p.K;
        p.OH = 1; p.OW = 1;
    }
    defer_hip::launch_conv_igemm(p, relu, (bool)res, gemm_mode, false, s);
    return out;
}

Tensor linear(Tensor x, Tensor w, c10::optional<Tensor> bias) {
    check_bf16(x, "x");
    check_bf16(w, "w");
    TORCH_CHECK(x.dim() == 2 && w.dim() == 2, "linear wants 2-D");
    int M = x.size(0), K = x.size(1), N = w.size(0);
    TORCH_CHECK(w.size(1) == K, "K mismatch");
    TORCH_CHECK(K % 8 == 0, "K must be a multiple of 8");
    auto out = at::empty({M, N}, x.options());
    c10::optional<Tensor> bias_f;
    if (bias)
        bias_f = cached_weight_prep(*bias, -1, [&] {
            return bias->to(at::kFloat).contiguous();
        });
    ConvParams p{};
    p.x = bptr(x); p.w = bptr(w);
    p.scale = fptr_opt(c10::nullopt, "scale", ones_buf(), N);
    p.bias = fptr_opt(bias_f, "bias", zeros_buf(), N);
    p.res = nullptr; p.out = bptr_mut(out); p.zbuf = zero_buf();
    p.M = M; p.K = K; p.Cout = N;
    p.NB = M; p.H = 1; p.W = 1; p.Cin = K;
    p.OH = 1; p.OW = 1; p.R = 1; p.S = 1; p.stride = 1; p.pad = 0;
    defer_hip::launch_conv_igemm(p, false, false, true, false, cur_stream());
    return out;
}

Tensor conv1x1_prebn(Tensor x, Tensor w, Tensor scale, Tensor bias,
                     c10::optional<Tensor> out_scale,
                     c10::optional<Tensor> out_bias) {
    // x [N,H,W,Cin] bf16, w [Cout,1,1,Cin] bf16, scale/bias fp32[Cin]:
    // out = relu(x*scale+bias) @ w — one pass over x instead of the
    // bn_act tensor round-trip (DenseNet's dominant cost)
    check_bf16(x, "x");
    check_bf16(w, "w");
    TORCH_CHECK(x.dim() == 4 && w.dim() == 4 && w.size(1) == 1
                && w.size(2) == 1, "conv1x1_prebn wants NHWC x, 1x1 w");
    int NB = x.size(0), H = x.size(1), W = x.size(2), Cin = x.size(3);
    int Cout = w.size(0);
    TORCH_CHECK(w.size(3) == Cin && Cin % 8 == 0, "Cin mismatch");
    TORCH_CHECK(scale.scalar_type() == at::kFloat && scale.is_contiguous()
                && scale.numel() == Cin, "scale must be fp32[Cin]");
    TORCH_CHECK(bias.scalar_type() == at::kFloat && bias.is_contiguous()
                && bias.numel() == Cin, "bias must be fp32[Cin]");
    long M = (long)NB * H * W;
    TORCH_CHECK(M < (1LL << 31), "M too large");
    auto out = at::empty({NB, H, W, Cout}, x.options());
    const float* os = nullptr;
    const float* ob = nullptr;
    if (out_scale) {
        TORCH_CHECK(out_bias, "out_scale needs out_bias");
        TORCH_CHECK(out_scale->scalar_type() == at::kFloat
                    && out_scale->is_contiguous()
                    && out_scale->numel() == Cout
                    && out_bias->scalar_type() == at::kFloat
                    && out_bias->is_contiguous()
                    && out_bias->numel() == Cout,
                    "out affine must be fp32[Cout]");
        os = out_scale->data_ptr<float>();
        ob = out_bias->data_ptr<float>();
    }
    defer_hip::launch_gemm_prebn(bptr(x), bptr(w),
                                 scale.data_ptr<float>(),
                                 bias.data_ptr<float>(), os, ob,
                                 zero_buf(), bptr_mut(out), (int)M, Cin,
                                 Cout, cur_stream());
    return out;
}

Tensor bn_act(Tensor x, Tensor scale, Tensor bias, bool relu) {
    check_bf16(x, "x");
    int C = x.size(-1);
    TORCH_CHECK(C % 8 == 0, "C must be a multiple of 8");
    auto out = at::empty_like(x);
    auto sf = scale.to(at::kFloat).contiguous();
    auto bf = bias.to(at::kFloat).contiguous();
    defer_hip::launch_bn_act(bptr(x), sf.data_ptr<float>(),
                             bf.data_ptr<float>(), bptr_mut(out),
                             x.numel() / 8, C / 8, relu, cur_stream());
    return out;
}

Tensor add_act(Tensor a, Tensor b, bool relu) {
    check_bf16(a, "a");
    check_bf16(b, "b");
    TORCH_CHECK(a.sizes() == b.sizes(), "shape mismatch");
    TORCH_CHECK(a.numel() % 8 == 0, "numel must be a multiple of 8");
    auto out = at::empty_like(a);
    defer_hip::launch_add_act(bptr(a), bptr(b), bptr_mut(out),
                              a.numel() / 8, relu, cur_stream());
    return out;
}

Tensor relu(Tensor x) {
    check_bf16(x, "x");
    TORCH_CHECK(x.numel() % 8 == 0, "numel must be a multiple of 8");
    auto out = at::empty_like(x);
    defer_hip::launch_relu(bptr(x), bptr_mut(out), x.numel() / 8,
                           cur_stream());
    return out;
}

Tensor softmax(Tensor x) {
    check_bf16(x, "x");
    int cols = x.size(-1);
    long rows = x.numel() / cols;
    auto out = at::empty_like(x);
    defer_hip::launch_softmax(bptr(x), bptr_mut(out), (int)rows, cols,
                              cur_stream());
    return out;
}

Tensor maxpool2d(Tensor x, int64_t kernel, int64_t stride, int64_t pad) {
    check_bf16(x, "x");
    TORCH_CHECK(x.dim() == 4, "x must be NHWC");
    int NB = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
    TORCH_CHECK(C % 8 == 0, "C must be a multiple of 8");
    int OH = (int)((H + 2 * pad - kernel) / stride + 1);
    int OW = (int)((W + 2 * pad - kernel) / stride + 1);
    auto out = at::empty({NB, OH, OW, C}, x.options());
    defer_hip::launch_maxpool(bptr(x), bptr_mut(out), NB, H, W, C, OH, OW,
                              (int)kernel, (int)stride, (int)pad,
                              cur_stream());
    return out;
}

Tensor avgpool2d(Tensor x, int64_t kernel, int64_t stride, int64_t pad) {
    check_bf16(x, "x");
    TORCH_CHECK(x.dim() == 4, "x must be NHWC");
    int NB = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
    TORCH_CHECK(C % 8 == 0, "C must be a multiple of 8");
    int OH = (int)((H + 2 * pad - kernel) / stride + 1);
    int OW = (int)((W + 2 * pad - kernel) / stride + 1);
    auto out = at::empty({NB, OH, OW, C}, x.options());
    defer_hip::launch_avgpool(bptr(x), bptr_mut(out), NB, H, W, C, OH,
                              OW, (int)kernel, (int)stride, (int)pad,
                              cur_stream());
    return out;
}

// Concat along the LAST (channel) dim of NHWC tensors: pure data
// movement, one strided device-to-device hipMemcpy2DAsync per input
// (row = one pixel's channels) — no torch compute kernels in the path
// (DenseNet's dense blocks are concat-heavy).
Tensor concat_lastdim(std::vector<Tensor> xs) {
    TORCH_CHECK(!xs.empty(), "concat of nothing");
    auto base = xs[0].sizes().vec();
    int nd = (int)base.size();
    long rows = 1;
    for (int i = 0; i + 1 < nd; ++i) rows *= base[i];
    long ctot = 0;
    for (auto& t : xs) {
        check_bf16(t, "concat input");
        TORCH_CHECK((int)t.sizes().size() == nd, "rank mismatch");
        for (int i = 0; i + 1 < nd; ++i)
            TORCH_CHECK(t.size(i) == base[i], "leading-dim mismatch");
        ctot += t.size(nd - 1);
    }
    base[nd - 1] = ctot;
    auto out = at::empty(base, xs[0].options());
    hipStream_t s = cur_stream();
    if (xs.size() == 2 && xs[0].size(nd - 1) % 8 == 0 &&
        xs[1].size(nd - 1) % 8 == 0) {
        // the DenseNet shape class: one vectorized kernel beats the
        // per-input small-pitch 2D copies (measured)
        defer_hip::launch_cat2(bptr(xs[0]), bptr(xs[1]), bptr_mut(out),
                               rows, (int)xs[0].size(nd - 1),
                               (int)xs[1].size(nd - 1), s);
        return out;
    }
    long coff = 0;
    for (auto& t : xs) {
        long cj = t.size(nd - 1);
        hipMemcpy2DAsync((char*)bptr_mut(out) + coff * 2, ctot * 2,
                         bptr(t), cj * 2, cj * 2, rows,
                         hipMemcpyDeviceToDevice, s);
        coff += cj;
    }
    return out;
}

Tensor global_avg_pool(Tensor x) {
    check_bf16(x, "x");
    TORCH_CHECK(x.dim() == 4, "x must be NHWC");
    int NB = x.size(0), HW = x.size(1) * x.size(2), C = x.size(3);
    TORCH_CHECK(C % 8 == 0, "C must be a multiple of 8");
    auto out = at::empty({NB, C}, x.options());
    defer_hip::launch_gap(bptr(x), bptr_mut(out), NB, HW, C, cur_stream());
    return out;
}

std::tuple<int64_t, int64_t, int64_t> field_shape(const Tensor& t) {
    auto sz = t.sizes();
    int nd = sz.size();
    if (nd == 1) return {1, 1, sz[0]};
    if (nd == 2) return {1, sz[0], sz[1]};
    int64_t d0 = 1;
    for (int i = 0; i + 2 < nd; ++i) d0 *= sz[i];
    return {d0, sz[nd - 2], sz[nd - 1]};
}

Tensor zfp_encode(Tensor x, int64_t rate, c10::optional<Tensor> out,
                  int64_t phases = 3) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x must be GPU contig");
    bool bf16 = x.scalar_type() == at::kBFloat16;
    TORCH_CHECK(bf16 || x.scalar_type() == at::kFloat, "bf16/f32 only");
    auto [d0, d1, d2] = field_shape(x);
    long nb = ((d0 + 3) / 4) * ((d1 + 3) / 4) * ((d2 + 3) / 4);
    long bytes = nb * rate * 8;
    Tensor o;
    if (out) {
        TORCH_CHECK(out->numel() == bytes && out->is_contiguous()
                    && out->scalar_type() == at::kByte, "bad out buffer");
        o = *out;
    } else {
        o = at::empty({bytes},
                      x.options().dtype(at::kByte));
    }
    defer_hip::launch_zfp_encode(bptr(x), bptr_mut(o), bf16, (int)d0,
                                 (int)d1, (int)d2, (int)rate,
                                 cur_stream(), (int)phases);
    return o;
}

Tensor zfp_decode(Tensor wire, std::vector<int64_t> shape, int64_t rate,
                  bool bf16_out) {
    TORCH_CHECK(wire.is_cuda() && wire.is_contiguous()
                && wire.scalar_type() == at::kByte, "bad wire");
    auto o = at::empty(shape, wire.options().dtype(
                                  bf16_out ? at::kBFloat16 : at::kFloat));
    auto [d0, d1, d2] = field_shape(o);
    defer_hip::launch_zfp_decode(bptr(wire), bptr_mut(o), bf16_out,
                                 (int)d0, (int)d1, (int)d2, (int)rate,
                                 cur_stream());
    return o;
}

// fp8 e4m3fn wire: one fused HIP path (init + atomic amax + cast, the
// scale never visits the host) replacing the torch fallback's ~5
// dispatches. Wire layout matches comm.Codec exactly.
void fp8_encode(Tensor x, Tensor out) {
    check_bf16(x, "x");
    long n = x.numel();
    TORCH_CHECK(out.is_cuda() && out.is_contiguous()
                && out.scalar_type() == at::kByte
                && out.numel() == n + 4, "bad fp8 wire buffer");
    defer_hip::launch_fp8_encode(bptr(x), n, bptr_mut(out),
                                 cur_stream());
}

Tensor fp8_decode(Tensor wire, std::vector<int64_t> shape) {
    TORCH_CHECK(wire.is_cuda() && wire.is_contiguous()
                && wire.scalar_type() == at::kByte, "bad wire");
    auto y = at::empty(shape, wire.options().dtype(at::kBFloat16));
    TORCH_CHECK(y.numel() + 4 == wire.numel(), "wire/shape mismatch");
    defer_hip::launch_fp8_decode(bptr(wire), y.numel(), bptr_mut(y),
                                 cur_stream());
    return y;
}

Tensor lz4_compress(Tensor x) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous()
                && x.scalar_type() == at::kByte, "x must be GPU u8 contig");
    long n = x.numel();
    TORCH_CHECK(n > 0, "empty input");
    auto scratch = at::empty({defer_hip::lz4_scratch_bytes(n)},
                             x.options());
    auto o = at::empty({defer_hip::lz4_max_compressed(n)}, x.options());
    defer_hip::launch_lz4_compress(bptr(x), n, bptr_mut(scratch),
                                   bptr_mut(o), cur_stream());
    // total compressed size = header word [2 + nblocks] (syncs the stream;
    // the hop needs the byte count on the host to post the send anyway)
    long nb = (n + 4095) / 4096;
    auto hdr_cpu = o.narrow(0, 4 * (2 + nb), 4).to(at::kCPU);
    uint32_t total;
    std::memcpy(&total, hdr_cpu.data_ptr(), 4);
    return o.narrow(0, 0, 4 * (2 + nb + 1) + (long)total);
}

// Fully async zfp+lz4 hop encode: compress `x` (u8, the ZFP wire) into
// the caller's preallocated worst-case `out` ring slot and write the
// total wire byte count into `len_out` (device int64) — no host sync.
// The hop sends `len_out` as the size message straight from the device
// and reads the host copy one item later (comm.VarP2PRing), keeping the
// sender's host loop ahead of the device (the round-1 version's
// per-item .to(kCPU) sync serialized the pipeline).
void lz4_compress_into(Tensor x, Tensor scratch, Tensor out,
                       Tensor len_out) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous()
                && x.scalar_type() == at::kByte, "x must be GPU u8 contig");
    long n = x.numel();
    TORCH_CHECK(n > 0, "empty input");
    TORCH_CHECK(scratch.numel() >= defer_hip::lz4_scratch_bytes(n)
                && scratch.is_cuda(), "scratch too small");
    TORCH_CHECK(out.numel() >= defer_hip::lz4_max_compressed(n)
                && out.is_cuda() && out.scalar_type() == at::kByte,
                "out too small");
    TORCH_CHECK(len_out.is_cuda() && len_out.numel() == 1
                && len_out.scalar_type() == at::kLong, "bad len_out");
    hipStream_t s = cur_stream();
    defer_hip::launch_lz4_compress(bptr(x), n, bptr_mut(scratch),
                                   bptr_mut(out), s);
    defer_hip::launch_lz4_wire_len(bptr(out), n, bptr_mut(len_out), s);
}

Tensor lz4_decompress(Tensor comp, int64_t raw_len) {
    TORCH_CHECK(comp.is_cuda() && comp.is_contiguous()
                && comp.scalar_type() == at::kByte, "bad comp buffer");
    TORCH_CHECK(raw_len > 0, "raw_len must be positive");
    auto o = at::empty({raw_len}, comp.options());
    defer_hip::launch_lz4_decompress(bptr(comp), bptr_mut(o), raw_len,
                                     cur_stream());
    return o;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("conv2d_bn_act", &conv2d_bn_act, py::arg("x"), py::arg("w"),
          py::arg("scale"), py::arg("bias"), py::arg("res"),
          py::arg("stride"), py::arg("pad"), py::arg("relu"));
    m.def("linear", &linear);
    m.def("bn_act", &bn_act);
    m.def("conv1x1_prebn", &conv1x1_prebn, py::arg("x"),
          py::arg("w"), py::arg("scale"), py::arg("bias"),
          py::arg("out_scale") = py::none(),
          py::arg("out_bias") = py::none());
    m.def("add_act", &add_act);
    m.def("relu", &relu);
    m.def("softmax", &softmax);
    m.def("maxpool2d", &maxpool2d);
    m.def("avgpool2d", &avgpool2d);
    m.def("concat_lastdim", &concat_lastdim);
    m.def("global_avg_pool", &global_avg_pool);
    m.def("zfp_encode", &zfp_encode, py::arg("x"), py::arg("rate"),
          py::arg("out") = py::none(), py::arg("phases") = 3);
    m.def("zfp_decode", &zfp_decode);
    m.def("fp8_encode", &fp8_encode);
    m.def("fp8_decode", &fp8_decode);
    m.def("lz4_compress", &lz4_compress);
    m.def("lz4_compress_into", &lz4_compress_into);
    m.def("lz4_scratch_bytes",
          [](int64_t n) { return defer_hip::lz4_scratch_bytes(n); });
    m.def("lz4_decompress", &lz4_decompress);
}
