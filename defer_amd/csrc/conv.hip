// Implicit-GEMM convolution for gfx950 (CDNA4), NHWC bf16, MFMA
// 16x16x32 with fp32 accumulation.
//
// GEMM view: C[M, N] = A[M, K] x B[K, N]
//   M = NB*OH*OW (output pixels), N = Cout, K = R*S*Cin,
//   A row m = input patch of pixel m (gathered on the fly),
//   B = weights, stored OHWI [Cout][R][S][Cin] so B^T rows are contiguous.
//
// Structure (cdna_hip_programming.md §5, T3+T4: counted vmcnt pipeline):
// 128x64 output tile, BK=64, 4 waves (each 32x64), TRIPLE-buffered LDS
// staged by global_load_lds width 16 (lane-linear dest; XOR swizzle
// applied to the *source* chunk index and the read address — rule 21).
// The stage cursor runs two K-tiles ahead; each iteration waits a COUNTED
// s_waitcnt vmcnt(OPS) (never 0 mid-loop) and a raw s_barrier, so glds
// stay in flight across barriers and the ~1 us HBM latency hides under
// two iterations of MFMA. Each block walks multiple m-tiles (grid-stride)
// so the pipeline never drains between tiles; when the whole K fits one
// tile (1x1 convs over <=64 input channels) the weight tile is staged
// once and kept resident in LDS (B_PERSIST).
//
// Epilogue fuses folded-BN scale/bias, residual add and ReLU (the Keras
// Conv2D+BN+Add+ReLU stack the reference executes via model.predict,
// /root/reference/src/node.py:106).
//
// GEMM mode (R=S=1, H=W=1, Cin=K): out[M,N] = x[M,K] @ w[N,K]^T — used
// for 1x1/s1 convs, the dense classifier head, and any explicit GEMM.
#include "common.h"
#include "kernels.h"

using defer_hip::ConvParams;

#define BM 128
#define BN 64
#define BK 64
#define NTHREADS 256
#define KCH (BK / 8)          // 16-B chunks per tile row (8)

// XOR swizzle: logical (row, k8) lives at physical k8p = k8 ^ (row & 7).
__device__ __forceinline__ int swz(int row, int k8) {
    return k8 ^ (row & 7);
}

// async 16B global->LDS (wave-uniform LDS base; HW writes lane i at
// base + i*16)
__device__ __forceinline__ void glds16(const bf16* src, bf16* lds_base) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)lds_base,
        16, 0, 0);
}

template <int ACT, bool HAS_RES, bool GEMM_MODE, bool B_PERSIST>
__global__ __launch_bounds__(NTHREADS, 2) void conv_igemm_kernel(
    ConvParams p) {
    const bf16* __restrict__ X = (const bf16*)p.x;
    const bf16* __restrict__ Wt = (const bf16*)p.w;
    const bf16* __restrict__ Z = (const bf16*)p.zbuf;
    const bf16* __restrict__ RES = (const bf16*)p.res;
    bf16* __restrict__ OUT = (bf16*)p.out;
    // LDS: A triple-buffer + B (single when persistent, triple otherwise)
    __shared__ __attribute__((aligned(16)))
    bf16 lds[(3 * BM + (B_PERSIST ? 1 : 3) * BN) * BK];
    bf16* A0 = lds;
    bf16* B0 = lds + 3 * BM * BK;

    const int tid = threadIdx.x;
    const int wave = tid / WAVE;
    const int lane = tid % WAVE;
    const int n0 = blockIdx.y * BN;
    const int mtiles = (p.M + BM - 1) / BM;
    const int nk = (p.K + BK - 1) / BK;

    // ---- per-thread chunk geometry (constant): 4 A chunks, 2 B chunks
    int a_row[4], a_k8[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        int chunk = wave * 256 + i * 64 + lane;
        a_row[i] = chunk / KCH;
        a_k8[i] = swz(a_row[i], chunk % KCH);
    }
    int b_row[2], b_k8[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
        int chunk = wave * 128 + i * 64 + lane;
        b_row[i] = chunk / KCH;
        b_k8[i] = swz(b_row[i], chunk % KCH);
    }

    const int lo16 = lane & 15;
    const int hi4 = lane >> 4;
    // folded-BN scale/bias for this block's 64 output channels, loaded
    // once (p.scale/p.bias are always non-null; the binding substitutes
    // cached ones/zeros) — per-element nullable loads inside the epilogue
    // made hipcc emit one vmcnt(0) per element (de-pipelining trap).
    const float* __restrict__ SCALE = p.scale;
    const float* __restrict__ BIAS = p.bias;
    float sc[4], bi[4];
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
        int n = n0 + ni * 16 + lo16;
        if (n >= p.Cout) n = p.Cout - 1;
        sc[ni] = SCALE[n];
        bi[ni] = BIAS[n];
    }

    auto stage_a = [&](int mt, int kt, int buf) {
        const int k0 = kt * BK;
        const int m0 = mt * BM;
        bf16* A = A0 + buf * BM * BK;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            int k = k0 + a_k8[i] * 8;
            int m = m0 + a_row[i];
            const bf16* src = Z;
            if (k < p.K && m < p.M) {
                if (GEMM_MODE) {
                    src = X + (long)m * p.K + k;
                } else {
                    int ow = m % p.OW;
                    int t = m / p.OW;
                    int oh = t % p.OH;
                    int nb = t / p.OH;
                    int c = k % p.Cin;
                    int rs = k / p.Cin;
                    int r = rs / p.S;
                    int s = rs % p.S;
                    int ih = oh * p.stride - p.pad + r;
                    int iw = ow * p.stride - p.pad + s;
                    if (ih >= 0 && ih < p.H && iw >= 0 && iw < p.W)
                        src = X + (((long)nb * p.H + ih) * p.W + iw)
                                    * p.Cin + c;
                }
            }
            glds16(src, A + (wave * 256 + i * 64) * 8);
        }
    };
    auto stage_b = [&](int kt, int buf) {
        const int k0 = kt * BK;
        bf16* B = B0 + buf * BN * BK;
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            int k = k0 + b_k8[i] * 8;
            int n = n0 + b_row[i];
            const bf16* src = (k < p.K && n < p.Cout)
                                  ? Wt + (long)n * p.K + k
                                  : Z;
            glds16(src, B + (wave * 128 + i * 64) * 8);
        }
    };

    f32x4 acc[2][4];
    auto compute = [&](int abuf, int bbuf) {
        bf16* A = A0 + abuf * BM * BK;
        bf16* B = B0 + bbuf * BN * BK;
#pragma unroll
        for (int ks = 0; ks < BK / 32; ++ks) {
            bf16x8 af[2], bfr[4];
#pragma unroll
            for (int mi = 0; mi < 2; ++mi) {
                int row = wave * 32 + mi * 16 + lo16;
                af[mi] = *reinterpret_cast<bf16x8*>(
                    A + row * BK + swz(row, ks * 4 + hi4) * 8);
            }
#pragma unroll
            for (int ni = 0; ni < 4; ++ni) {
                int row = ni * 16 + lo16;
                bfr[ni] = *reinterpret_cast<bf16x8*>(
                    B + row * BK + swz(row, ks * 4 + hi4) * 8);
            }
#pragma unroll
            for (int mi = 0; mi < 2; ++mi)
#pragma unroll
                for (int ni = 0; ni < 4; ++ni)
                    acc[mi][ni] =
                        MFMA_BF16_16x16x32(af[mi], bfr[ni], acc[mi][ni]);
        }
    };

    auto epilogue = [&](int mt, int abuf) {
        // Coalesced epilogue via an LDS transpose bounce through the A
        // buffer just consumed (free until stage((it+2)%3), which differs
        // from abuf): the MFMA C-fragment layout is column-scattered, so
        // direct stores are 32 scalar 2-B ops per lane whose completion
        // the next counted vmcnt would drain. Instead: acc*scale+bias ->
        // XOR-swizzled f32 LDS tile (two 64-row halves), then every wave
        // reads rows back contiguously and does 8-B residual loads +
        // 8-B bf16 stores.
        float* scratch = (float*)(A0 + abuf * BM * BK);   // 16 KB
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();       // all waves done with A[abuf]
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            if ((wave >> 1) == h) {
                const int rbase = (wave & 1) * 32;
#pragma unroll
                for (int mi = 0; mi < 2; ++mi)
#pragma unroll
                    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
                        for (int e = 0; e < 4; ++e) {
                            int r = rbase + mi * 16 + hi4 * 4 + e;
                            int c = ni * 16 + lo16;
                            int cs = c ^ (((r >> 2) & 3) << 4);
                            scratch[r * 64 + cs] =
                                acc[mi][ni][e] * sc[ni] + bi[ni];
                        }
            }
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();
            // readback: 64 rows x 16 chunks of 4 floats, 4 chunks/thread
            // Interior blocks (the common case) take a branch-free
            // vector path: batch-issue 4 LDS reads + 4 residual loads,
            // sched_barrier, then convert+store — so hipcc emits counted
            // waits instead of one vmcnt(0) per chunk (de-pipelining
            // trap) and no per-chunk exec-mask dances.
            const bool interior =
                (mt * BM + BM <= p.M) && (n0 + BN <= p.Cout);
            if (interior) {
                f32x4 v4[4];
                bf16x4 rv[4];
                long off[4];
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    int chunk = tid + i * 256;
                    int r = chunk >> 4;
                    int c4 = (chunk & 15) * 4;
                    int cs = c4 ^ (((r >> 2) & 3) << 4);
                    off[i] = (long)(mt * BM + h * 64 + r) * p.Cout
                             + n0 + c4;
                    v4[i] = *reinterpret_cast<f32x4*>(
                        scratch + r * 64 + cs);
                    if (HAS_RES)
                        rv[i] = *reinterpret_cast<const bf16x4*>(
                            RES + off[i]);
                }
                __builtin_amdgcn_sched_barrier(0);
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    bf16x4 o;
#pragma unroll
                    for (int j = 0; j < 4; ++j) {
                        float v = v4[i][j];
                        if (HAS_RES) v += bf2f(rv[i][j]);
                        o[j] = f2bf(apply_act(v, ACT));
                    }
                    *reinterpret_cast<bf16x4*>(OUT + off[i]) = o;
                }
            } else {
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    int chunk = tid + i * 256;
                    int r = chunk >> 4;
                    int c4 = (chunk & 15) * 4;
                    int cs = c4 ^ (((r >> 2) & 3) << 4);
                    int m = mt * BM + h * 64 + r;
                    int n = n0 + c4;
                    if (m >= p.M || n >= p.Cout) continue;
                    f32x4 v4 = *reinterpret_cast<f32x4*>(
                        scratch + r * 64 + cs);
                    long off = (long)m * p.Cout + n;
#pragma unroll
                    for (int j = 0; j < 4; ++j) {
                        if (n + j >= p.Cout) continue;
                        float v = v4[j];
                        if (HAS_RES) v += bf2f(RES[off + j]);
                        OUT[off + j] = f2bf(apply_act(v, ACT));
                    }
                }
            }
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();   // next half reuses scratch
        }
    };

    // ---- flattened (m-tile, k-tile) pipeline with a 2-tile-deep stage
    // cursor. Per-wave glds ops per staged tile: 4 A (+2 B unless
    // persistent). vmcnt waits are counted so prefetches stay in flight.
    constexpr int OPS = B_PERSIST ? 4 : 6;

    int s_mt = blockIdx.x, s_kt = 0;
    auto advance = [&]() {
        if (++s_kt == nk) { s_kt = 0; s_mt += gridDim.x; }
    };
    auto stage = [&](int buf) {
        stage_a(s_mt, s_kt, buf);
        if (!B_PERSIST) stage_b(s_kt, buf);
        advance();
    };

    if (B_PERSIST) stage_b(0, 0);   // oldest ops: drained by first wait
    int staged = 0;
    if (s_mt < mtiles) { stage(0); ++staged; }
    if (s_mt < mtiles) { stage(1); ++staged; }

    int it = 0;   // computed-iteration counter (tile i lives in buf i%3)
    for (int mt = blockIdx.x; mt < mtiles; mt += gridDim.x) {
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
#pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};
        for (int kt = 0; kt < nk; ++kt, ++it) {
            // tile `it` landed when <= OPS*(tiles-in-flight-behind-it)
            // ops remain outstanding
            if (staged - it - 1 >= 1)
                asm volatile("s_waitcnt vmcnt(%0)" ::"i"(OPS) : "memory");
            else
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();
            compute(it % 3, B_PERSIST ? 0 : it % 3);
            if (kt == nk - 1) epilogue(mt, it % 3);
            // refill: buffer (it+2)%3 was last read at compute(it-1),
            // which every wave finished before this iteration's barrier
            if (s_mt < mtiles) { stage((it + 2) % 3); ++staged; }
        }
    }
}

// ---------------------------------------------------------------------------
// Channel pad, NHWC: [rows, C] -> [rows, C8] zero-padded (stem Cin=3 -> 8;
// also pads OHWI weights viewed as [Cout*R*S, Cin]).
__global__ void pad_channels_kernel(const bf16* __restrict__ x,
                                    bf16* __restrict__ y, long rows, int C,
                                    int C8) {
    long total = rows * (C8 / 8);
    long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long gs = (long)gridDim.x * blockDim.x;
    for (long i = i0; i < total; i += gs) {
        int cb = (int)(i % (C8 / 8));
        long row = i / (C8 / 8);
        bf16x8 v;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            int c = cb * 8 + j;
            v[j] = (c < C) ? x[row * C + c] : (bf16)0.f;
        }
        store_bf16x8(y + row * C8 + cb * 8, v);
    }
}

// ---------------------------------------------------------------------------
// host launchers
namespace defer_hip {

void launch_conv_igemm(const ConvParams& p, bool relu, bool has_res,
                       bool gemm_mode, hipStream_t s) {
    const int mtiles = (p.M + BM - 1) / BM;
    const int ny = (p.Cout + BN - 1) / BN;
    // target ~3 blocks/CU x 256 CUs; each block m-loops the rest
    int gx = mtiles;
    const int target = 768;
    if ((long)mtiles * ny > target) {
        gx = target / ny > 0 ? target / ny : 1;
        if (gx > mtiles) gx = mtiles;
    }
    dim3 grid(gx, ny);
    dim3 block(NTHREADS);
    const int nk = (p.K + BK - 1) / BK;
    const bool bp = (nk == 1);
#define DISPATCH4(A, R, G, BP) \
    hipLaunchKernelGGL((conv_igemm_kernel<A, R, G, BP>), grid, block, 0, \
                       s, p)
#define DISPATCH2(A, R)                                        \
    do {                                                       \
        if (gemm_mode) {                                       \
            if (bp) DISPATCH4(A, R, true, true);               \
            else DISPATCH4(A, R, true, false);                 \
        } else {                                               \
            if (bp) DISPATCH4(A, R, false, true);              \
            else DISPATCH4(A, R, false, false);                \
        }                                                      \
    } while (0)
    if (relu) { if (has_res) DISPATCH2(ACT_RELU, true);
                else DISPATCH2(ACT_RELU, false); }
    else      { if (has_res) DISPATCH2(ACT_NONE, true);
                else DISPATCH2(ACT_NONE, false); }
#undef DISPATCH2
#undef DISPATCH4
}

static int grid1d(long work, int block) {
    long g = (work + block - 1) / block;
    return (int)(g < 2048 ? g : 2048);
}

void launch_pad_channels(const void* x, void* y, long rows, int C, int C8,
                         hipStream_t s) {
    hipLaunchKernelGGL(pad_channels_kernel,
                       dim3(grid1d(rows * (C8 / 8), 256)), dim3(256), 0, s,
                       (const bf16*)x, (bf16*)y, rows, C, C8);
}

}  // namespace defer_hip
