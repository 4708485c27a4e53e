// Fused pre-activation 1x1 convolution for gfx950: out = (relu(x*s+b)) @ W^T
// with the per-INPUT-channel folded BN applied while staging x — the
// DenseNet composite (BN -> ReLU -> 1x1 conv, models/layers.py
// BNActConv). The standalone bn_act pass over the growing concat tensors
// was 57% of the DenseNet step (profiles/densenet121_b128_kernel_stats
// .csv): fusing it removes one full read+write pass of the layer input.
//
// Deliberately a SEPARATE kernel from conv_igemm_kernel: the tuned igemm
// template stages A directly global->LDS (counted-vmcnt pipeline), which
// cannot apply per-element math; this kernel stages A through registers
// (load -> fp32 bn+relu -> bf16 -> ds_write, same XOR-swizzled layout)
// with synchronous per-tile barriers and relies on high block residency
// (16 KiB LDS, launch_bounds 4 blocks/CU) for latency cover. k-tiles
// accumulate in the same ascending order with the same 16x16x32 MFMA
// fragment mapping as the igemm GEMM mode, so the fused result is
// BIT-IDENTICAL to bn_act -> conv2d_bn_act (tests/test_ops_gpu.py).
#include "common.h"
#include "kernels.h"

#define PB_BK 64
#define PB_TPB 256

__device__ __forceinline__ int pb_swz(int row, int k8) {
    return k8 ^ (row & 7);
}

__device__ __forceinline__ void pb_glds16(const bf16* src,
                                          bf16* lds_base) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)lds_base,
        16, 0, 0);
}

// OUT_AFF: apply a per-OUTPUT-channel affine + ReLU on the fp32
// accumulator before rounding (the NEXT layer's folded BN-ReLU — in a
// DenseNet dense layer, c1's output feeds only norm2/relu2/conv2, so
// norm2 folds here and conv2 becomes a plain 3x3). Applied on fp32 acc
// (no intermediate bf16 round), so it is slightly MORE accurate than
// the two-step path, not bit-identical to it.
template <int BM, int BN, bool OUT_AFF>
__global__ __launch_bounds__(PB_TPB, 4) void gemm_prebn_kernel(
    const bf16* __restrict__ X,      // [M][K]
    const bf16* __restrict__ Wt,     // [Cout][K]
    const float* __restrict__ PS,    // [K] folded BN scale
    const float* __restrict__ PBb,   // [K] folded BN bias
    const float* __restrict__ OS,    // [Cout] out scale (OUT_AFF)
    const float* __restrict__ OB,    // [Cout] out bias (OUT_AFF)
    const bf16* __restrict__ Z,      // >=16B zeros
    bf16* __restrict__ OUT,          // [M][Cout]
    int M, int K, int Cout) {
    constexpr int KCH = PB_BK / 8;
    constexpr int NW = PB_TPB / WAVE;
    constexpr int WMW = BM / 32;
    constexpr int WNW = NW / WMW;
    constexpr int WN = BN / WNW;
    constexpr int NI = WN / 16;
    constexpr int ACH = BM * 8 / PB_TPB;
    constexpr int BCH = BN * 8 / PB_TPB;

    __shared__ __attribute__((aligned(16))) bf16 lds[(BM + BN) * PB_BK];
    bf16* A0 = lds;
    bf16* B0 = lds + BM * PB_BK;

    const int tid = threadIdx.x;
    const int wave = tid / WAVE;
    const int lane = tid % WAVE;
    const int wm = wave / WNW;
    const int wn = wave % WNW;
    const int n0 = blockIdx.y * BN;
    const int lo16 = lane & 15;
    const int hi4 = lane >> 4;
    const int mtiles = (M + BM - 1) / BM;
    const int nk = (K + PB_BK - 1) / PB_BK;

    int a_row[ACH], a_k8[ACH];
#pragma unroll
    for (int i = 0; i < ACH; ++i) {
        int chunk = wave * (ACH * 64) + i * 64 + lane;
        a_row[i] = chunk / KCH;
        a_k8[i] = pb_swz(a_row[i], chunk % KCH);
    }
    int b_row[BCH], b_k8[BCH];
#pragma unroll
    for (int i = 0; i < BCH; ++i) {
        int chunk = wave * (BCH * 64) + i * 64 + lane;
        b_row[i] = chunk / KCH;
        b_k8[i] = pb_swz(b_row[i], chunk % KCH);
    }

    for (int mt = blockIdx.x; mt < mtiles; mt += gridDim.x) {
        const int m0 = mt * BM;
        f32x4 acc[2][NI];
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
#pragma unroll
            for (int ni = 0; ni < NI; ++ni)
                acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

        for (int kt = 0; kt < nk; ++kt) {
            // ---- B: direct global -> LDS (weights, no transform)
#pragma unroll
            for (int i = 0; i < BCH; ++i) {
                int n = n0 + b_row[i];
                int k = kt * PB_BK + b_k8[i] * 8;
                const bf16* src = (n < Cout && k < K)
                                      ? Wt + (long)n * K + k
                                      : Z;
                pb_glds16(src, B0 + (wave * (BCH * 64) + i * 64) * 8);
            }
            // ---- A: registers, bn+relu in fp32, bf16 round, ds_write
            // (same rounding as the standalone bn_act kernel -> the
            // fused path is bit-identical to the two-step one)
#pragma unroll
            for (int i = 0; i < ACH; ++i) {
                int m = m0 + a_row[i];
                int k = kt * PB_BK + a_k8[i] * 8;
                bf16x8 o;
                if (m < M && k < K) {
                    bf16x8 v = load_bf16x8(X + (long)m * K + k);
                    f32x4 s0 = *reinterpret_cast<const f32x4*>(PS + k);
                    f32x4 s1 = *reinterpret_cast<const f32x4*>(PS + k
                                                               + 4);
                    f32x4 c0 = *reinterpret_cast<const f32x4*>(PBb + k);
                    f32x4 c1 = *reinterpret_cast<const f32x4*>(PBb + k
                                                               + 4);
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        float sj = j < 4 ? s0[j] : s1[j - 4];
                        float cj = j < 4 ? c0[j] : c1[j - 4];
                        o[j] = f2bf(fmaxf(fmaf(bf2f(v[j]), sj, cj),
                                          0.f));
                    }
                } else {
#pragma unroll
                    for (int j = 0; j < 8; ++j) o[j] = f2bf(0.f);
                }
                store_bf16x8(A0 + (wave * (ACH * 64) + i * 64) * 8
                                 + lane * 8, o);
            }
            asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();
            // ---- compute (same fragment mapping + k order as igemm)
#pragma unroll
            for (int ks = 0; ks < PB_BK / 32; ++ks) {
                bf16x8 af[2], bfr[NI];
#pragma unroll
                for (int mi = 0; mi < 2; ++mi) {
                    int row = wm * 32 + mi * 16 + lo16;
                    af[mi] = *reinterpret_cast<bf16x8*>(
                        A0 + row * PB_BK
                           + pb_swz(row, ks * 4 + hi4) * 8);
                }
#pragma unroll
                for (int ni = 0; ni < NI; ++ni) {
                    int row = wn * WN + ni * 16 + lo16;
                    bfr[ni] = *reinterpret_cast<bf16x8*>(
                        B0 + row * PB_BK
                           + pb_swz(row, ks * 4 + hi4) * 8);
                }
#pragma unroll
                for (int mi = 0; mi < 2; ++mi)
#pragma unroll
                    for (int ni = 0; ni < NI; ++ni)
                        acc[mi][ni] = MFMA_BF16_16x16x32(af[mi], bfr[ni],
                                                         acc[mi][ni]);
            }
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();   // LDS reused next k-tile
        }
        // ---- store (direct; the outputs here are small vs the fused
        // input pass this kernel saves)
        float osc[NI], obi[NI];
        if (OUT_AFF)
#pragma unroll
            for (int ni = 0; ni < NI; ++ni) {
                int n = n0 + wn * WN + ni * 16 + lo16;
                if (n >= Cout) n = Cout - 1;
                osc[ni] = OS[n];
                obi[ni] = OB[n];
            }
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
#pragma unroll
            for (int ni = 0; ni < NI; ++ni)
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    int m = m0 + wm * 32 + mi * 16 + hi4 * 4 + e;
                    int n = n0 + wn * WN + ni * 16 + lo16;
                    float v = acc[mi][ni][e];
                    if (OUT_AFF)
                        v = fmaxf(fmaf(v, osc[ni], obi[ni]), 0.f);
                    if (m < M && n < Cout)
                        OUT[(long)m * Cout + n] = f2bf(v);
                }
    }
}

namespace defer_hip {

void launch_gemm_prebn(const void* x, const void* w, const float* ps,
                       const float* pb, const float* os, const float* ob,
                       const void* zbuf, void* out, int M, int K,
                       int Cout, hipStream_t s) {
    constexpr int BM = 64, BN = 64;
    int mtiles = (M + BM - 1) / BM;
    int ntiles = (Cout + BN - 1) / BN;
    int gx = mtiles < 4096 ? mtiles : 4096;
    if (os != nullptr)
        hipLaunchKernelGGL((gemm_prebn_kernel<BM, BN, true>),
                           dim3(gx, ntiles), dim3(PB_TPB), 0, s,
                           (const bf16*)x, (const bf16*)w, ps, pb, os,
                           ob, (const bf16*)zbuf, (bf16*)out, M, K,
                           Cout);
    else
        hipLaunchKernelGGL((gemm_prebn_kernel<BM, BN, false>),
                           dim3(gx, ntiles), dim3(PB_TPB), 0, s,
                           (const bf16*)x, (const bf16*)w, ps, pb, os,
                           ob, (const bf16*)zbuf, (bf16*)out, M, K,
                           Cout);
}

}  // namespace defer_hip
