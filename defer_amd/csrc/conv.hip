// Implicit-GEMM convolution for gfx950 (CDNA4), NHWC bf16, MFMA
// 16x16x32 with fp32 accumulation.
//
// GEMM view: C[M, N] = A[M, K] x B[K, N]
//   M = NB*OH*OW (output pixels), N = Cout, K = R*S*Cin,
//   A row m = input patch of pixel m (gathered on the fly),
//   B = weights, stored OHWI [Cout][R][S][Cin] so B^T rows are contiguous.
//
// Structure (cdna_hip_programming.md §5: the "step-3" / minimum 2-phase
// shape): 128x64 output tile, BK=64, 4 waves (each 32x64), double-buffered
// LDS staged by global_load_lds width 16 (lane-linear dest, XOR swizzle
// applied to the *source* chunk index and the read address — rule 21),
// one vmcnt(0)+barrier per K-tile. Epilogue fuses folded-BN scale/bias,
// residual add and ReLU (the Keras Conv2D+BN+Add+ReLU stack the reference
// executes via model.predict, /root/reference/src/node.py:106).
//
// The same kernel is the dense/GEMM path: R=S=1, H=W=1, Cin=K gives
// out[M,N] = x[M,K] @ w[N,K]^T (+bias, act) — used for the classifier
// head and the im2col'd stem conv.
#include "common.h"
#include "kernels.h"

using defer_hip::ConvParams;

#define BM 128
#define BN 64
#define BK 64
#define NTHREADS 256
// chunks are 16-byte (8 bf16) units; A tile = BM*BK bf16 = 1024 chunks,
// B tile = BN*BK = 512 chunks. Per wave: A 4 glds issues, B 2.
#define A_CHUNKS (BM * BK / 8)
#define B_CHUNKS (BN * BK / 8)
#define KCH (BK / 8)          // chunks per row (8)

// XOR swizzle: logical (row, k8) lives at physical k8p = k8 ^ (row & 7).
// Read side applies the same XOR on the byte address.
__device__ __forceinline__ int swz(int row, int k8) {
    return k8 ^ (row & 7);
}

// async 16B global->LDS for one lane's chunk (wave-uniform LDS base;
// hardware writes lane i at base + i*16)
__device__ __forceinline__ void glds16(const bf16* src, bf16* lds_base) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)lds_base,
        16, 0, 0);
}

template <int ACT, bool HAS_RES, bool GEMM_MODE>
__global__ __launch_bounds__(NTHREADS, 2) void conv_igemm_kernel(
    ConvParams p) {
    const bf16* __restrict__ X = (const bf16*)p.x;
    const bf16* __restrict__ Wt = (const bf16*)p.w;
    const bf16* __restrict__ Z = (const bf16*)p.zbuf;
    const bf16* __restrict__ RES = (const bf16*)p.res;
    bf16* __restrict__ OUT = (bf16*)p.out;
    __shared__ __attribute__((aligned(16))) bf16 lds[2 * (BM + BN) * BK];
    bf16* A0 = lds;                         // [BM][BK] x2
    bf16* B0 = lds + 2 * BM * BK;           // [BN][BK] x2

    const int tid = threadIdx.x;
    const int wave = tid / WAVE;
    const int lane = tid % WAVE;
    const int m0 = blockIdx.x * BM;
    const int n0 = blockIdx.y * BN;

    // ---- per-thread A-source precomputation (constant across K-tiles).
    // This wave's glds lanes cover physical chunks (wave*256 + i*64 + lane)
    // of the A tile; chunk -> (row m, physical k8) -> logical k8.
    int a_row[4];        // tile-local row of each of my 4 A chunks
    int a_k8[4];         // logical k8 (after inverse swizzle)
    const bf16* a_base[4];  // pixel base pointer (at ih0, iw0, c=0)
    int a_ih0[4], a_iw0[4];
    bool a_mvalid[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        int chunk = wave * 256 + i * 64 + lane;   // physical chunk index
        int row = chunk / KCH;
        int k8p = chunk % KCH;
        a_row[i] = row;
        a_k8[i] = swz(row, k8p);                  // logical k8
        int m = m0 + row;
        a_mvalid[i] = (m < p.M);
        int mm = a_mvalid[i] ? m : 0;
        if (GEMM_MODE) {
            a_base[i] = X + (long)mm * p.K;
            a_ih0[i] = 0; a_iw0[i] = 0;
        } else {
            int ow = mm % p.OW;
            int t = mm / p.OW;
            int oh = t % p.OH;
            int nb = t / p.OH;
            int ih0 = oh * p.stride - p.pad;
            int iw0 = ow * p.stride - p.pad;
            a_ih0[i] = ih0; a_iw0[i] = iw0;
            a_base[i] = X + (((long)nb * p.H + ih0) * p.W + iw0) * p.Cin;
        }
    }
    // B chunks: physical chunks (wave*128 + i*64 + lane)
    int b_row[2], b_k8[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
        int chunk = wave * 128 + i * 64 + lane;
        b_row[i] = chunk / KCH;
        b_k8[i] = swz(b_row[i], chunk % KCH);
    }

    const int nk = (p.K + BK - 1) / BK;

    // ---- staging: issue glds for K-tile kt into buffer buf (0/1)
    auto stage = [&](int kt, int buf) {
        const int k0 = kt * BK;
        bf16* A = A0 + buf * BM * BK;
        bf16* B = B0 + buf * BN * BK;
        // A gather
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            int k = k0 + a_k8[i] * 8;
            const bf16* src = Z;
            if (k < p.K && a_mvalid[i]) {
                if (GEMM_MODE) {
                    src = a_base[i] + k;
                } else {
                    int c = k % p.Cin;
                    int rs = k / p.Cin;
                    int r = rs / p.S;
                    int s = rs % p.S;
                    int ih = a_ih0[i] + r;
                    int iw = a_iw0[i] + s;
                    if (ih >= 0 && ih < p.H && iw >= 0 && iw < p.W)
                        src = a_base[i] + ((long)r * p.W + s) * p.Cin + c;
                }
            }
            // wave-uniform LDS base for this glds issue
            glds16(src, A + (wave * 256 + i * 64) * 8);
        }
        // B weights: row n = n0 + b_row, contiguous in k
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

    // ---- MFMA compute on buffer buf
    f32x4 acc[2][4];
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

    const int lo16 = lane & 15;
    const int hi4 = lane >> 4;   // 0..3
    auto compute = [&](int buf) {
        bf16* A = A0 + buf * BM * BK;
        bf16* B = B0 + buf * BN * BK;
#pragma unroll
        for (int ks = 0; ks < BK / 32; ++ks) {
            const int kk = ks * 32 + hi4 * 8;   // bf16 index in row
            const int k8 = kk / 8;
            bf16x8 af[2], bf[4];
#pragma unroll
            for (int mi = 0; mi < 2; ++mi) {
                int row = wave * 32 + mi * 16 + lo16;
                af[mi] = *reinterpret_cast<bf16x8*>(
                    A + row * BK + swz(row, k8) * 8);
            }
#pragma unroll
            for (int ni = 0; ni < 4; ++ni) {
                int row = ni * 16 + lo16;
                bf[ni] = *reinterpret_cast<bf16x8*>(
                    B + row * BK + swz(row, k8) * 8);
            }
#pragma unroll
            for (int mi = 0; mi < 2; ++mi)
#pragma unroll
                for (int ni = 0; ni < 4; ++ni)
                    acc[mi][ni] =
                        MFMA_BF16_16x16x32(af[mi], bf[ni], acc[mi][ni]);
        }
    };

    // ---- main loop: minimum 2-phase (stage t+1 before compute t)
    stage(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    int cur = 0;
    for (int kt = 0; kt < nk - 1; ++kt) {
        stage(kt + 1, cur ^ 1);
        compute(cur);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __syncthreads();
        cur ^= 1;
    }
    compute(cur);

    // ---- epilogue: scale/bias + residual + act, bf16 stores
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
}

// ---------------------------------------------------------------------------
// im2col for the stem conv (Cin=3: channel chunks are too thin for the
// implicit gather). Produces [M][K_pad] bf16, zero-padded beyond
// K = R*S*Cin, which then runs through the GEMM mode of the kernel above.
__global__ void im2col_kernel(const bf16* __restrict__ x, bf16* __restrict__ out,
                              int NB, int H, int W, int Cin,
                              int OH, int OW, int R, int S,
                              int stride, int pad, int Kpad) {
    long M = (long)NB * OH * OW;
    int K = R * S * Cin;
    long total = M * Kpad;
    long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long gs = (long)gridDim.x * blockDim.x;
    for (long i = i0; i < total; i += gs) {
        int k = (int)(i % Kpad);
        long m = i / Kpad;
        bf16 v = (bf16)0.f;
        if (k < K) {
            int c = k % Cin;
            int rs = k / Cin;
            int r = rs / S, s = rs % S;
            int ow = (int)(m % OW);
            long t = m / OW;
            int oh = (int)(t % OH);
            int nb = (int)(t / OH);
            int ih = oh * stride - pad + r;
            int iw = ow * stride - pad + s;
            if (ih >= 0 && ih < H && iw >= 0 && iw < W)
                v = x[(((long)nb * H + ih) * W + iw) * Cin + c];
        }
        out[i] = v;
    }
}

// Zero-pad a weight tensor's K dim: [Cout][K] -> [Cout][Kpad]
__global__ void padk_kernel(const bf16* __restrict__ w, bf16* __restrict__ out,
                            int Cout, int K, int Kpad) {
    long total = (long)Cout * Kpad;
    long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long gs = (long)gridDim.x * blockDim.x;
    for (long i = i0; i < total; i += gs) {
        int k = (int)(i % Kpad);
        long n = i / Kpad;
        out[i] = (k < K) ? w[n * K + k] : (bf16)0.f;
    }
}

// ---------------------------------------------------------------------------
// host launchers
namespace defer_hip {

void launch_conv_igemm(const ConvParams& p, bool relu, bool has_res,
                       bool gemm_mode, hipStream_t s) {
    dim3 grid((p.M + BM - 1) / BM, (p.Cout + BN - 1) / BN);
    dim3 block(NTHREADS);
#define DISPATCH(A, R, G) \
    hipLaunchKernelGGL((conv_igemm_kernel<A, R, G>), grid, block, 0, s, p)
    if (gemm_mode) {
        if (relu) { if (has_res) DISPATCH(ACT_RELU, true, true);
                    else DISPATCH(ACT_RELU, false, true); }
        else      { if (has_res) DISPATCH(ACT_NONE, true, true);
                    else DISPATCH(ACT_NONE, false, true); }
    } else {
        if (relu) { if (has_res) DISPATCH(ACT_RELU, true, false);
                    else DISPATCH(ACT_RELU, false, false); }
        else      { if (has_res) DISPATCH(ACT_NONE, true, false);
                    else DISPATCH(ACT_NONE, false, false); }
    }
#undef DISPATCH
}

static int grid1d(long work, int block) {
    long g = (work + block - 1) / block;
    return (int)(g < 2048 ? g : 2048);
}

void launch_im2col(const void* x, void* out, int NB, int H, int W, int Cin,
                   int OH, int OW, int R, int S, int stride, int pad,
                   int Kpad, hipStream_t s) {
    long M = (long)NB * OH * OW;
    hipLaunchKernelGGL(im2col_kernel, dim3(grid1d(M * Kpad, 256)),
                       dim3(256), 0, s, (const bf16*)x, (bf16*)out, NB, H,
                       W, Cin, OH, OW, R, S, stride, pad, Kpad);
}

void launch_padk(const void* w, void* out, int Cout, int K, int Kpad,
                 hipStream_t s) {
    hipLaunchKernelGGL(padk_kernel, dim3(grid1d((long)Cout * Kpad, 256)),
                       dim3(256), 0, s, (const bf16*)w, (bf16*)out, Cout, K,
                       Kpad);
}

}  // namespace defer_hip
