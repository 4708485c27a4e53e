// Implicit-GEMM convolution for gfx950 (CDNA4), NHWC bf16, MFMA
// 16x16x32 with fp32 accumulation.
//
// GEMM view: C[M, N] = A[M, K] x B[K, N]
//   M = NB*OH*OW (output pixels), N = Cout, K = R*S*Cin,
//   A row m = input patch of pixel m (gathered on the fly),
//   B = weights, stored OHWI [Cout][R][S][Cin] so B^T rows are contiguous.
//
// Structure (cdna_hip_programming.md §5, "minimum 2-phase" + m-tile loop):
// 128x64 output tile, BK=64, 4 waves (each 32x64), double-buffered LDS
// staged by global_load_lds width 16 (lane-linear dest; XOR swizzle
// applied to the *source* chunk index and the read address — rule 21),
// one vmcnt(0)+barrier per K-tile. Each block walks multiple m-tiles
// (grid-stride) so the pipeline never drains between tiles and — when the
// whole K fits one tile (1x1 convs over <=64 input channels) — the weight
// tile is staged once and kept resident in LDS (B_PERSIST).
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
__global__ __launch_bounds__(NTHREADS, 3) void conv_igemm_kernel(
    ConvParams p) {
    const bf16* __restrict__ X = (const bf16*)p.x;
    const bf16* __restrict__ Wt = (const bf16*)p.w;
    const bf16* __restrict__ Z = (const bf16*)p.zbuf;
    const bf16* __restrict__ RES = (const bf16*)p.res;
    bf16* __restrict__ OUT = (bf16*)p.out;
    // LDS: A double-buffer + B (single when persistent, double otherwise)
    __shared__ __attribute__((aligned(16)))
    bf16 lds[(2 * BM + (B_PERSIST ? 1 : 2) * BN) * BK];
    bf16* A0 = lds;
    bf16* B0 = lds + 2 * BM * BK;

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
    const int lo16 = lane & 15;
    const int hi4 = lane >> 4;
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

    auto epilogue = [&](int mt) {
        const int m0 = mt * BM;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
            int n = n0 + ni * 16 + lo16;
            if (n >= p.Cout) continue;
            float sc = p.scale ? p.scale[n] : 1.0f;
            float bi = p.bias ? p.bias[n] : 0.0f;
#pragma unroll
            for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    int m = m0 + wave * 32 + mi * 16 + hi4 * 4 + e;
                    if (m >= p.M) continue;
                    float v = acc[mi][ni][e] * sc + bi;
                    if (HAS_RES) v += bf2f(RES[(long)m * p.Cout + n]);
                    OUT[(long)m * p.Cout + n] = f2bf(apply_act(v, ACT));
                }
            }
        }
    };

    // ---- flattened (m-tile, k-tile) pipeline; stage cursor runs one
    // iteration ahead of the compute cursor.
    int s_mt = blockIdx.x, s_kt = 0;
    stage_a(s_mt, s_kt, 0);
    stage_b(s_kt, 0);
    // advance stage cursor
    if (++s_kt == nk) { s_kt = 0; s_mt += gridDim.x; }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

    int cur = 0;
    for (int mt = blockIdx.x; mt < mtiles; mt += gridDim.x) {
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
#pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};
        for (int kt = 0; kt < nk; ++kt) {
            if (s_mt < mtiles) {
                stage_a(s_mt, s_kt, cur ^ 1);
                if (!B_PERSIST) stage_b(s_kt, cur ^ 1);
                if (++s_kt == nk) { s_kt = 0; s_mt += gridDim.x; }
            }
            compute(cur, B_PERSIST ? 0 : cur);
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __syncthreads();
            cur ^= 1;
        }
        epilogue(mt);
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
