// Fixed-rate ZFP-style activation codec for gfx950 — the MI355X-native
// rebuild of the reference's zfpy compression of boundary activations
// (/root/reference/src/dispatcher.py:81-84, node.py:107).
//
// Bit-exact to the numpy spec in defer_amd/ops/zfp_ref.py. Wave-native
// design: ONE 64-lane wavefront owns one 4x4x4 block — value <-> lane,
// the integer Haar lifting runs on cross-lane shuffles, a bit-plane word
// is one __ballot, and lane 0 serializes the group-tested bit stream.
// Blocks are independent and fixed-size (rate*8 bytes), so encode and
// decode are embarrassingly parallel across the tensor.
#include <type_traits>

#include "common.h"
#include "kernels.h"

#define QBITS 25
#define PLANES 30
#define HDR_BITS 16
#define NBMASK 0xAAAAAAAAu

// sequency permutation: PERM[i] = coefficient position stored at stream
// slot i (sorted by i+j+k, then (i,j,k)); IPERM is its inverse.
__constant__ unsigned char ZPERM[64] = {
     0,  1,  4, 16,  2,  5,  8, 17, 20, 32,  3,  6,  9, 12, 18, 21,
    24, 33, 36, 48,  7, 10, 13, 19, 22, 25, 28, 34, 37, 40, 49, 52,
    11, 14, 23, 26, 29, 35, 38, 41, 44, 50, 53, 56, 15, 27, 30, 39,
    42, 45, 51, 54, 57, 60, 31, 43, 46, 55, 58, 61, 47, 59, 62, 63
};
__constant__ unsigned char ZIPERM[64] = {
     0,  1,  4, 10,  2,  5, 11, 20,  6, 12, 21, 32, 13, 22, 33, 44,
     3,  7, 14, 23,  8, 15, 24, 34, 16, 25, 35, 45, 26, 36, 46, 54,
     9, 17, 27, 37, 18, 28, 38, 47, 29, 39, 48, 55, 40, 49, 56, 60,
    19, 30, 41, 50, 31, 42, 51, 57, 43, 52, 58, 61, 53, 59, 62, 63
};

// ---- exact integer Haar lifting on 4-lane groups (stride s) -------------
__device__ __forceinline__ int fwd_axis(int q, int lane, int s) {
    int r = (lane / s) & 3;
    // step 1: pairs (0,1), (2,3): even -> l=(a+b)>>1, odd -> h=a-b
    int o = __shfl_xor(q, s);
    q = ((r & 1) == 0) ? ((q + o) >> 1) : (o - q);
    // step 2: positions 0,2: even -> ll=(l0+l1)>>1, r==2 -> hl=l0-l1
    o = __shfl_xor(q, 2 * s);
    if ((r & 1) == 0) q = (r == 0) ? ((q + o) >> 1) : (o - q);
    // remap (ll, h0, hl, h1) -> (ll, hl, h0, h1): swap positions 1,2
    int base = lane - r * s;
    static const int MAPV[4] = {0, 2, 1, 3};
    return __shfl(q, base + MAPV[r] * s);
}

__device__ __forceinline__ int inv_axis(int q, int lane, int s) {
    int r = (lane / s) & 3;
    // un-remap: swap positions 1,2 -> (ll, h0, hl, h1)
    int base = lane - r * s;
    static const int MAPV[4] = {0, 2, 1, 3};
    q = __shfl(q, base + MAPV[r] * s);
    // undo step 2 (positions 0 (ll) and 2 (hl)):
    //   l1 = ll - (hl>>1); l0 = l1 + hl
    int o = __shfl_xor(q, 2 * s);
    if ((r & 1) == 0)
        q = (r == 0) ? ((q - (o >> 1)) + o) : (o - (q >> 1));
    // undo step 1: a1 = l - (h>>1); a0 = a1 + h
    o = __shfl_xor(q, s);
    q = ((r & 1) == 0) ? ((q - (o >> 1)) + o) : (o - (q >> 1));
    return q;
}

// ---- wave bit-plane transpose -------------------------------------------
// Replaces the encoder's 30x __ballot plane-collect loop and the
// decoder's 30x ds_read_b64 gather (measured ~85% of the transform
// phase). The matrix is 64 values x 30
// planes, so 32-bit rows suffice — five u32 butterfly steps transpose
// the two 32x32 halves (lane-xor j < 32 stays inside each half) and one
// cross-half shuffle stitches the u64 plane words. Roughly half the
// VALU ops of the u64 version (the transform phase is issue-bound).
__device__ __forceinline__ u32 bt32_steps(u32 x, int lane) {
    static const u32 M5[5] = {0x0000FFFFu, 0x00FF00FFu, 0x0F0F0F0Fu,
                              0x33333333u, 0x55555555u};
#pragma unroll
    for (int i = 0; i < 5; ++i) {
        const int j = 16 >> i;
        const u32 m = M5[i];
        u32 y = __shfl_xor(x, j);
        if ((lane & j) == 0)
            x ^= (((x >> j) ^ y) & m) << j;
        else
            x ^= ((y >> j) ^ x) & m;
    }
    return x;
}

// encode: lane L holds u32 plane-bits of value L -> lanes p < 32 hold
// the u64 word of plane p (bit L = value L's bit p); lanes >= 32 return
// garbage (only lanes < PLANES store).
__device__ __forceinline__ u64 planes_from_values(u32 u, int lane) {
    u32 x = bt32_steps(u, lane);
    u32 hi = __shfl_xor(x, 32);      // lane l < 32 reads lane l+32
    return (u64)x | ((u64)hi << 32);
}

// decode: lanes p < 32 hold plane word w_p (lanes 30/31 zero) -> every
// lane L gets its value's u32 plane-bits.
__device__ __forceinline__ u32 values_from_planes(u64 w, int lane) {
    u64 other = __shfl_xor(w, 32);
    u32 x = (lane < 32) ? (u32)w : (u32)(other >> 32);
    return bt32_steps(x, lane);
}

// ---- two-block interleaved forms ----------------------------------------
// The transform phase is 60% SQ_WAIT at 2 waves/SIMD (LDS-capped
// occupancy; profiles/zfp_codec_throughput.txt PMC): the cross-lane
// chain's latency is exposed. Convergent ops execute in program order,
// so the only way to overlap two blocks' chains is to interleave the
// statements by hand — each helper advances two independent chains
// alternately, halving exposed latency per block pair.
template <int NC>
__device__ __forceinline__ void fwd_axisN(int q[NC], int lane, int s) {
    int r = (lane / s) & 3;
    int o[NC];
#pragma unroll
    for (int c = 0; c < NC; ++c) o[c] = __shfl_xor(q[c], s);
#pragma unroll
    for (int c = 0; c < NC; ++c)
        q[c] = ((r & 1) == 0) ? ((q[c] + o[c]) >> 1) : (o[c] - q[c]);
#pragma unroll
    for (int c = 0; c < NC; ++c) o[c] = __shfl_xor(q[c], 2 * s);
#pragma unroll
    for (int c = 0; c < NC; ++c)
        if ((r & 1) == 0)
            q[c] = (r == 0) ? ((q[c] + o[c]) >> 1) : (o[c] - q[c]);
    static const int MAPV[4] = {0, 2, 1, 3};
    int dst = lane - r * s + MAPV[r] * s;
#pragma unroll
    for (int c = 0; c < NC; ++c) q[c] = __shfl(q[c], dst);
}

template <int NC>
__device__ __forceinline__ void planes_from_valuesN(u32 u[NC], int lane,
                                                    u64 w[NC]) {
    static const u32 M5[5] = {0x0000FFFFu, 0x00FF00FFu, 0x0F0F0F0Fu,
                              0x33333333u, 0x55555555u};
#pragma unroll
    for (int i = 0; i < 5; ++i) {
        const int j = 16 >> i;
        const u32 m = M5[i];
        u32 y[NC];
#pragma unroll
        for (int c = 0; c < NC; ++c) y[c] = __shfl_xor(u[c], j);
#pragma unroll
        for (int c = 0; c < NC; ++c) {
            if ((lane & j) == 0)
                u[c] ^= (((u[c] >> j) ^ y[c]) & m) << j;
            else
                u[c] ^= ((y[c] >> j) ^ u[c]) & m;
        }
    }
#pragma unroll
    for (int c = 0; c < NC; ++c) {
        u32 h = __shfl_xor(u[c], 32);
        w[c] = (u64)u[c] | ((u64)h << 32);
    }
}

template <int NC>
__device__ __forceinline__ void inv_axisN(int q[NC], int lane, int s) {
    int r = (lane / s) & 3;
    static const int MAPV[4] = {0, 2, 1, 3};
    int dst = lane - r * s + MAPV[r] * s;
#pragma unroll
    for (int c = 0; c < NC; ++c) q[c] = __shfl(q[c], dst);
    int o[NC];
#pragma unroll
    for (int c = 0; c < NC; ++c) o[c] = __shfl_xor(q[c], 2 * s);
#pragma unroll
    for (int c = 0; c < NC; ++c)
        if ((r & 1) == 0)
            q[c] = (r == 0) ? ((q[c] - (o[c] >> 1)) + o[c])
                            : (o[c] - (q[c] >> 1));
#pragma unroll
    for (int c = 0; c < NC; ++c) o[c] = __shfl_xor(q[c], s);
#pragma unroll
    for (int c = 0; c < NC; ++c)
        q[c] = ((r & 1) == 0) ? ((q[c] - (o[c] >> 1)) + o[c])
                              : (o[c] - (q[c] >> 1));
}

template <int NC>
__device__ __forceinline__ void values_from_planesN(u64 w[NC], int lane,
                                                    u32 u[NC]) {
    static const u32 M5[5] = {0x0000FFFFu, 0x00FF00FFu, 0x0F0F0F0Fu,
                              0x33333333u, 0x55555555u};
    u64 other[NC];
#pragma unroll
    for (int c = 0; c < NC; ++c) other[c] = __shfl_xor(w[c], 32);
#pragma unroll
    for (int c = 0; c < NC; ++c)
        u[c] = (lane < 32) ? (u32)w[c] : (u32)(other[c] >> 32);
#pragma unroll
    for (int i = 0; i < 5; ++i) {
        const int j = 16 >> i;
        const u32 m = M5[i];
        u32 y[NC];
#pragma unroll
        for (int c = 0; c < NC; ++c) y[c] = __shfl_xor(u[c], j);
#pragma unroll
        for (int c = 0; c < NC; ++c) {
            if ((lane & j) == 0)
                u[c] ^= (((u[c] >> j) ^ y[c]) & m) << j;
            else
                u[c] ^= ((y[c] >> j) ^ u[c]) & m;
        }
    }
}

// ---- lane-0 bit stream ---------------------------------------------------
struct BitWriter {
    u32* out;       // current word pointer
    u32 acc;
    int nacc;       // bits in acc
    int budget;     // remaining bits

    __device__ void put1(int bit) {
        if (budget <= 0) return;
        --budget;
        acc |= (u32)bit << nacc;
        if (++nacc == 32) { *out++ = acc; acc = 0; nacc = 0; }
    }
    __device__ void put_bits(u64 w, int nb) {
        if (nb > budget) nb = budget;
        budget -= nb;
        while (nb > 0) {
            int take = 32 - nacc;
            if (take > nb) take = nb;
            u32 mask = (take < 32) ? ((1u << take) - 1u) : 0xffffffffu;
            acc |= ((u32)w & mask) << nacc;
            w >>= take;
            nacc += take;
            nb -= take;
            if (nacc == 32) { *out++ = acc; acc = 0; nacc = 0; }
        }
    }
    __device__ void finish(u32* end) {   // flush + zero-fill fixed size
        if (nacc) { *out++ = acc; acc = 0; nacc = 0; }
        while (out < end) *out++ = 0;
    }
};

// Word-window bit reader: keeps up to 64 stream bits in `win`; take(k)
// returns the next k bits LSB-first (zeros past the end of the stream,
// like the reference's exhausted reads).
struct BitReader {
    const u32* in;
    u64 win;
    int nwin;        // bits in window
    int left;        // stream bits not yet in window

    __device__ void init(const u32* p, int bits) {
        in = p; win = 0; nwin = 0; left = bits;
        refill();
    }
    __device__ void refill() {
        while (nwin <= 32 && left > 0) {
            u32 w = *in++;
            int t = left < 32 ? left : 32;
            u64 ww = (t < 32) ? (w & ((1u << t) - 1u)) : w;
            win |= ww << nwin;
            nwin += t;
            left -= t;
        }
    }
    __device__ u64 take(int k) {   // k <= 64; zeros past stream end
        u64 v = 0;
        int got = 0;
        while (got < k) {
            refill();
            if (nwin == 0) break;
            int t = (k - got) < nwin ? (k - got) : nwin;
            u64 mask = (t >= 64) ? ~0ull : ((1ull << t) - 1ull);
            v |= (win & mask) << got;
            win = (t >= 64) ? 0 : (win >> t);
            nwin -= t;
            got += t;
        }
        return v;
    }
    __device__ bool empty() const { return nwin == 0 && left == 0; }
};

// ---------------------------------------------------------------------------
// Two-phase encode (one kernel, LDS hand-off). The bit stream of a block
// is inherently serial, so phase 1 (wave-per-block: gather, lift, plane
// ballots -> LDS) runs at wave parallelism and phase 2 gives every
// THREAD one whole block to serialize from its LDS plane words — 256
// serializers per workgroup instead of one lane per wave. Bit-exact to
// the single-lane version: the emission code is identical, only who
// executes it changed.
#define CGRP 256                     // blocks per workgroup round

// Panel staging (PANEL=true, requires b2 % 16 == 0 so a 16-block run
// shares one (bi,bj) row tile): the per-lane direct gather reads 16
// scattered 8-byte segments per block — ~1/8 of each cache line used,
// which capped encode at ~130 GB/s (profiles/zfp_codec_throughput.txt).
// Instead each wave owns 64 CONSECUTIVE blocks per round, staged as 4
// panels of 16 blocks: a panel's footprint is 16 rows x 64 contiguous
// d2-values, loaded with fully dense 128-byte row reads into an LDS
// slab and re-read block-wise (row stride padded to dodge bank
// conflicts). All boundary tensors the pipeline ships have C >= 64, so
// the panel path is the hot one; b2 % 16 != 0 falls back to the
// direct-gather path.
template <bool BF16_IN, int PH = 3,   // PH: bit0=transform, bit1=serialize
          bool PANEL = false>
__global__ __launch_bounds__(CGRP, 2) void zfp_encode_kernel(
    const void* __restrict__ xv, u32* __restrict__ out, int d0, int d1,
    int d2, int b0, int b1, int b2, int rate) {
    const int tid = threadIdx.x;
    const int lane = tid % WAVE;
    const int wavei = tid / WAVE;
    const int nwaves = blockDim.x / WAVE;
    const long nblocks = (long)b0 * b1 * b2;
    const int li = lane >> 4, lj = (lane >> 2) & 3, lk = lane & 3;
    const int wpb = rate * 2;                   // u32 words per block

    __shared__ u64 s_planes[CGRP][PLANES + 1];  // +1: LDS bank pad
    __shared__ u32 s_hdr[CGRP];                 // 0 = zero block
    using elem_t = typename std::conditional<BF16_IN, bf16, float>::type;
    constexpr int SLAB_W = BF16_IN ? 66 : 65;   // pad: rows spread banks
    __shared__ elem_t s_slab[4][16][PANEL ? SLAB_W : 1];

    // incremental block coordinates for this wave's stride-nwaves walk
    // (no runtime division in the loop) and a PF-deep gather pipeline so
    // one block's load latency hides under the previous blocks' work
    constexpr int PF = 4;
    for (long base = (long)blockIdx.x * CGRP; base < nblocks;
         base += (long)gridDim.x * CGRP) {
        // ---- phase 1 (panel): 4 panels of 16 consecutive blocks/wave
        if (PANEL && (PH & 1)) {
            for (int p = 0; p < 64 / 16; ++p) {
                long pb = base + wavei * 64 + p * 16;
                if (pb >= nblocks) break;
                int npan = (int)(nblocks - pb < 16 ? nblocks - pb : 16);
                int bk0 = (int)(pb % b2);
                long t = pb / b2;
                int bj = (int)(t % b1), bi = (int)(t / b1);
                // dense row reads: 64 contiguous values per row
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    int gi = bi * 4 + (r >> 2); if (gi > d0 - 1) gi = d0 - 1;
                    int gj = bj * 4 + (r & 3);  if (gj > d1 - 1) gj = d1 - 1;
                    int gk = bk0 * 4 + lane;    if (gk > d2 - 1) gk = d2 - 1;
                    long idx = ((long)gi * d1 + gj) * d2 + gk;
                    s_slab[wavei][r][lane] =
                        ((const elem_t*)xv)[idx];
                }
                // hand-interleaved block PAIRS (PH 1/3): two
                // independent cross-lane chains advanced alternately —
                // convergent ops execute in program order, so source
                // interleave is the only way to overlap the chains'
                // LDS-pipe latency (60% SQ_WAIT at 2 waves/SIMD). Zero
                // blocks waste a transform (uniform-rare in activation
                // data) but stay bit-exact — planes just not written.
                if (PH != 4 && PH != 5) {
                    constexpr int NC = 8;
                    int b = 0;
                    for (; b + NC - 1 < npan; b += NC) {
                        float v[NC], a[NC];
#pragma unroll
                        for (int c = 0; c < NC; ++c) {
                            elem_t raw = s_slab[wavei][lane >> 2]
                                               [(b + c) * 4 + (lane & 3)];
                            v[c] = BF16_IN ? bf2f(*(bf16*)&raw)
                                           : *(float*)&raw;
                            a[c] = fabsf(v[c]);
                        }
#pragma unroll
                        for (int off = 32; off > 0; off >>= 1)
#pragma unroll
                            for (int c = 0; c < NC; ++c)
                                a[c] = fmaxf(a[c],
                                             __shfl_xor(a[c], off));
                        bool act[NC];
                        int e[NC], q[NC];
#pragma unroll
                        for (int c = 0; c < NC; ++c) {
                            act[c] = (a[c] > 0.f) && isfinite(a[c]);
                            frexpf(a[c], &e[c]);
                            q[c] = (int)rintf(
                                v[c] * ldexpf(1.0f, QBITS - e[c]));
                        }
                        fwd_axisN<NC>(q, lane, 1);
                        fwd_axisN<NC>(q, lane, 4);
                        fwd_axisN<NC>(q, lane, 16);
                        u32 u[NC];
#pragma unroll
                        for (int c = 0; c < NC; ++c) {
                            q[c] = __shfl(q[c], ZPERM[lane]);
                            u[c] = ((u32)q[c] + NBMASK) ^ NBMASK;
                        }
                        if (lane == 0)
#pragma unroll
                            for (int c = 0; c < NC; ++c)
                                s_hdr[wavei * 64 + p * 16 + b + c] =
                                    act[c]
                                    ? ((1u << 15)
                                       | ((u32)(e[c] + 256) & 0x1FFu))
                                    : 0u;
                        u64 w[NC];
                        planes_from_valuesN<NC>(u, lane, w);
#pragma unroll
                        for (int c = 0; c < NC; ++c)
                            if (act[c] && lane < PLANES)
                                s_planes[wavei * 64 + p * 16 + b + c]
                                        [lane] = w[c];
                    }
                    for (; b < npan; ++b) {   // tail (unreachable under
                        int s = wavei * 64 + p * 16 + b;  // PANEL align)
                        elem_t raw = s_slab[wavei][lane >> 2]
                                           [b * 4 + (lane & 3)];
                        float v = BF16_IN ? bf2f(*(bf16*)&raw)
                                          : *(float*)&raw;
                        float av = fabsf(v);
#pragma unroll
                        for (int off = 32; off > 0; off >>= 1)
                            av = fmaxf(av, __shfl_xor(av, off));
                        const bool active = (av > 0.f) && isfinite(av);
                        int emax;
                        frexpf(av, &emax);
                        int q = (int)rintf(v * ldexpf(1.0f,
                                                      QBITS - emax));
                        q = fwd_axis(q, lane, 1);
                        q = fwd_axis(q, lane, 4);
                        q = fwd_axis(q, lane, 16);
                        q = __shfl(q, ZPERM[lane]);
                        u32 u = ((u32)q + NBMASK) ^ NBMASK;
                        if (lane == 0)
                            s_hdr[s] = active
                                ? ((1u << 15)
                                   | ((u32)(emax + 256) & 0x1FFu)) : 0u;
                        u64 myw = planes_from_values(u, lane);
                        if (active && lane < PLANES)
                            s_planes[s][lane] = myw;
                    }
                    continue;           // next panel
                }
                // scalar loop: perf-bisect probes (PH 4/5) only
                for (int b = 0; b < npan; ++b) {
                    int s = wavei * 64 + p * 16 + b;
                    elem_t raw = s_slab[wavei][lane >> 2]
                                       [b * 4 + (lane & 3)];
                    float v = BF16_IN ? bf2f(*(bf16*)&raw)
                                      : *(float*)&raw;
                    float av = fabsf(v);
#pragma unroll
                    for (int off = 32; off > 0; off >>= 1)
                        av = fmaxf(av, __shfl_xor(av, off));
                    if (PH == 4) {       // perf bisect: stage+reduce only
                        if (av == 12345.678f) s_hdr[s] = 1;
                        continue;
                    }
                    const bool active = (av > 0.f) && isfinite(av);
                    int emax;
                    frexpf(av, &emax);
                    int q = (int)rintf(v * ldexpf(1.0f, QBITS - emax));
                    q = fwd_axis(q, lane, 1);
                    q = fwd_axis(q, lane, 4);
                    q = fwd_axis(q, lane, 16);
                    q = __shfl(q, ZPERM[lane]);
                    u32 u = ((u32)q + NBMASK) ^ NBMASK;
                    if (PH == 5) {       // perf bisect: + lift, no ballots
                        if (u == 0xdeadbeefu) s_hdr[s] = u;
                        continue;
                    }
                    if (lane == 0)
                        s_hdr[s] = active
                            ? ((1u << 15) | ((u32)(emax + 256) & 0x1FFu))
                            : 0u;
                    // butterfly transpose: lane p ends with plane p's
                    // 64-value word (one coalesced 30-lane ds_write)
                    u64 myw = planes_from_values(u, lane);
                    if (active && lane < PLANES) s_planes[s][lane] = myw;
                }
            }
        }
        // ---- phase 1 (direct gather): each wave strides CGRP/nwaves
        if (!PANEL && (PH & 1)) {
            long blk0 = base + wavei;
            int bk = (int)(blk0 % b2);
            long t = blk0 / b2;
            int bj = (int)(t % b1);
            int bi = (int)(t / b1);
            // raw-typed prefetch ring: the convert happens at USE so the
            // compiler leaves the loads in flight (counted waits)
            bf16 braw[PF];
            float fraw[PF];
            auto issue = [&](int d) {           // load block slot d ahead
                int gi = bi * 4 + li; if (gi > d0 - 1) gi = d0 - 1;
                int gj = bj * 4 + lj; if (gj > d1 - 1) gj = d1 - 1;
                int gk = bk * 4 + lk; if (gk > d2 - 1) gk = d2 - 1;
                long idx = ((long)gi * d1 + gj) * d2 + gk;
                if (BF16_IN)
                    braw[d] = ((const bf16*)xv)[idx];
                else
                    fraw[d] = ((const float*)xv)[idx];
                bk += nwaves;                   // advance block coords
                while (bk >= b2) { bk -= b2; ++bj; }
                while (bj >= b1) { bj -= b1; ++bi; }
            };
            // blocks this wave owns in the round
            long avail = nblocks - (base + wavei);
            int total = avail <= 0 ? 0
                        : (int)((avail < CGRP - wavei ? avail
                                                      : CGRP - wavei)
                                + nwaves - 1) / nwaves;
#pragma unroll
            for (int d = 0; d < PF; ++d)
                if (d < total) issue(d);
            for (int ii = 0; ii < total; ++ii) {
                int s = wavei + ii * nwaves;
                float v = BF16_IN ? bf2f(braw[ii % PF])
                                  : fraw[ii % PF];
                if (ii + PF < total) issue((ii + PF) % PF);
                float av = fabsf(v);
#pragma unroll
                for (int off = 32; off > 0; off >>= 1)
                    av = fmaxf(av, __shfl_xor(av, off));
                if (!(av > 0.f) || !isfinite(av)) {
                    if (lane == 0) s_hdr[s] = 0;
                    continue;
                }
                int emax;
                frexpf(av, &emax);
                // quantize (f32 product, round-half-even = numpy rint)
                int q = (int)rintf(v * ldexpf(1.0f, QBITS - emax));
                q = fwd_axis(q, lane, 1);
                q = fwd_axis(q, lane, 4);
                q = fwd_axis(q, lane, 16);
                q = __shfl(q, ZPERM[lane]);
                u32 u = ((u32)q + NBMASK) ^ NBMASK;
                if (lane == 0)
                    s_hdr[s] = (1u << 15) | ((u32)(emax + 256) & 0x1FFu);
                // butterfly transpose: lane p ends with plane p's word;
                // ONE coalesced 30-lane ds_write (the 30x __ballot loop
                // this replaces was ~85% of the transform phase, and
                // before that 30 single-lane stores serialized it on
                // the LDS pipe — both measured, profiles/README.md)
                u64 myw = planes_from_values(u, lane);
                if (lane < PLANES) s_planes[s][lane] = myw;
            }
        }
        __syncthreads();
        // ---- phase 2: thread tid serializes block base+tid
        long blk = base + tid;
        if ((PH & 2) && blk < nblocks) {
            u32* bout = out + blk * wpb;
            u32 hdr = s_hdr[tid];
            if (hdr == 0) {
                for (int w = 0; w < wpb; ++w) bout[w] = 0;
            } else {
                BitWriter wr{bout, 0u, 0, rate * 64};
                wr.put_bits(hdr, HDR_BITS);
                int n = 0;
                for (int p = PLANES - 1; p >= 0 && wr.budget > 0; --p) {
                    u64 x = s_planes[tid][p];
                    // significant prefix
                    wr.put_bits(x & ((n < 64) ? ((1ull << n) - 1ull)
                                              : ~0ull), n);
                    x >>= n;
                    // group tests, run-wise (bit-identical to per-bit
                    // emission; put_bits truncates at the budget)
                    while (n < 64) {
                        if (wr.budget <= 0) break;
                        int has = (x != 0);
                        wr.put1(has);
                        if (!has) break;
                        int tz = __builtin_ctzll(x);
                        int emit = tz + 1;
                        wr.put_bits(1ull << tz, emit);
                        x = (emit >= 64) ? 0 : (x >> emit);
                        n += emit;
                    }
                }
                wr.finish(bout + wpb);
            }
        }
        __syncthreads();               // LDS reused next round
    }
}

// Two-phase decode, mirror of the encoder: phase 1 gives every THREAD
// one block's bit stream to parse into LDS plane words (256 parallel
// parsers); phase 2 runs the inverse transform wave-per-block from LDS.
// Truncation endgame differs from the per-bit reference only in zero
// bits / internal n, so reconstruction is identical.
template <bool BF16_OUT, bool PANEL = false>
__global__ __launch_bounds__(CGRP, 2) void zfp_decode_kernel(
    const u32* __restrict__ wire, void* __restrict__ yv, int d0, int d1,
    int d2, int b0, int b1, int b2, int rate) {
    const int tid = threadIdx.x;
    const int lane = tid % WAVE;
    const int wavei = tid / WAVE;
    const int nwaves = blockDim.x / WAVE;
    const long nblocks = (long)b0 * b1 * b2;
    const int li = lane >> 4, lj = (lane >> 2) & 3, lk = lane & 3;
    const int wpb = rate * 2;

    __shared__ u64 s_planes[CGRP][PLANES + 1];  // +1: LDS bank pad
    __shared__ u32 s_hdr[CGRP];
    using elem_t = typename std::conditional<BF16_OUT, bf16, float>::type;
    constexpr int SLAB_W = BF16_OUT ? 66 : 65;
    __shared__ elem_t s_slab[4][16][PANEL ? SLAB_W : 1];

    for (long base = (long)blockIdx.x * CGRP; base < nblocks;
         base += (long)gridDim.x * CGRP) {
        // ---- phase 1: thread tid parses block base+tid
        long pblk = base + tid;
        if (pblk < nblocks) {
            BitReader rd;
            rd.init(wire + pblk * wpb, rate * 64);
            u32 hdr = (u32)rd.take(HDR_BITS);
            s_hdr[tid] = hdr;
            if (hdr >> 15) {
                int n = 0;
                for (int p = PLANES - 1; p >= 0; --p) {
                    u64 x = 0;
                    if (rd.empty()) {
                        // remaining planes are zero
                        for (int pp = p; pp >= 0; --pp)
                            s_planes[tid][pp] = 0;
                        break;
                    }
                    x = rd.take(n);                   // significant prefix
                    while (n < 64) {                  // group runs
                        if (rd.empty()) break;
                        if (!rd.take(1)) break;       // group test
                        // scan zeros (may span window refills) until the
                        // run's 1 bit
                        bool found = false;
                        while (n < 64) {
                            rd.refill();
                            int lim = 64 - n;
                            int avail = rd.nwin < lim ? rd.nwin : lim;
                            if (avail == 0) break;    // stream end
                            u64 w = rd.win &
                                    ((avail >= 64)
                                         ? ~0ull
                                         : ((1ull << avail) - 1ull));
                            int tz = w ? __builtin_ctzll(w) : avail;
                            if (tz < avail) {
                                rd.take(tz + 1);
                                x |= 1ull << (n + tz);
                                n += tz + 1;
                                found = true;
                                break;
                            }
                            rd.take(avail);           // all zeros
                            n += avail;
                            if (rd.empty()) break;
                        }
                        if (!found) break;
                    }
                    s_planes[tid][p] = x;
                }
            }
        }
        __syncthreads();
        // ---- phase 2 (panel): inverse of the encoder's staging — the
        // wave reconstructs 16 consecutive blocks into the LDS slab,
        // then writes 16 dense 64-value rows (the direct per-lane
        // scatter wastes ~7/8 of every written cache line)
        if (PANEL) {
            for (int p = 0; p < 64 / 16; ++p) {
                long pb = base + wavei * 64 + p * 16;
                if (pb >= nblocks) break;
                int npan = (int)(nblocks - pb < 16 ? nblocks - pb : 16);
                int bk0 = (int)(pb % b2);
                long t = pb / b2;
                int bj = (int)(t % b1), bi = (int)(t / b1);
                // four inverse chains hand-interleaved per wave (same
                // latency argument as the encoder); dead-header blocks
                // compute garbage that the flag masks to 0
                constexpr int NC = 8;
                int b = 0;
                for (; b + NC - 1 < npan; b += NC) {
                    u64 w[NC];
                    u32 u[NC];
                    int q[NC], em[NC];
                    bool live[NC];
#pragma unroll
                    for (int c = 0; c < NC; ++c) {
                        int s = wavei * 64 + p * 16 + b + c;
                        u32 hdr = s_hdr[s];
                        live[c] = (hdr >> 15) != 0;
                        em[c] = (int)(hdr & 0x1FFu) - 256;
                        w[c] = (lane < PLANES) ? s_planes[s][lane] : 0;
                    }
                    values_from_planesN<NC>(w, lane, u);
#pragma unroll
                    for (int c = 0; c < NC; ++c)
                        q[c] = (int)((u[c] ^ NBMASK) - NBMASK);
#pragma unroll
                    for (int c = 0; c < NC; ++c)
                        q[c] = __shfl(q[c], ZIPERM[lane]);
                    inv_axisN<NC>(q, lane, 16);
                    inv_axisN<NC>(q, lane, 4);
                    inv_axisN<NC>(q, lane, 1);
#pragma unroll
                    for (int c = 0; c < NC; ++c) {
                        float outv = live[c]
                            ? ldexpf((float)q[c], em[c] - QBITS) : 0.f;
                        if (BF16_OUT) {
                            bf16 h = f2bf(outv);
                            s_slab[wavei][lane >> 2]
                                  [(b + c) * 4 + (lane & 3)] =
                                *(elem_t*)&h;
                        } else {
                            s_slab[wavei][lane >> 2]
                                  [(b + c) * 4 + (lane & 3)] =
                                *(elem_t*)&outv;
                        }
                    }
                }
                for (; b < npan; ++b) {   // tail (unreachable: PANEL
                    int s = wavei * 64 + p * 16 + b;    // keeps npan 16)
                    u32 hdr = s_hdr[s];
                    float outv = 0.f;
                    if (hdr >> 15) {
                        int emax = (int)(hdr & 0x1FFu) - 256;
                        u64 w = (lane < PLANES) ? s_planes[s][lane] : 0;
                        u32 u = values_from_planes(w, lane);
                        int q = (int)((u ^ NBMASK) - NBMASK);
                        q = __shfl(q, ZIPERM[lane]);
                        q = inv_axis(q, lane, 16);
                        q = inv_axis(q, lane, 4);
                        q = inv_axis(q, lane, 1);
                        outv = ldexpf((float)q, emax - QBITS);
                    }
                    if (BF16_OUT) {
                        bf16 h = f2bf(outv);
                        s_slab[wavei][lane >> 2][b * 4 + (lane & 3)] =
                            *(elem_t*)&h;
                    } else {
                        s_slab[wavei][lane >> 2][b * 4 + (lane & 3)] =
                            *(elem_t*)&outv;
                    }
                }
                // dense row writes (partial rows predicated per lane)
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    int gi = bi * 4 + (r >> 2);
                    int gj = bj * 4 + (r & 3);
                    if (gi >= d0 || gj >= d1) continue;
                    int gk = bk0 * 4 + lane;
                    if (gk >= d2 || gk >= (bk0 + npan) * 4) continue;
                    long idx = ((long)gi * d1 + gj) * d2 + gk;
                    ((elem_t*)yv)[idx] = s_slab[wavei][r][lane];
                }
            }
            __syncthreads();           // LDS reused next round
            continue;
        }
        // ---- phase 2 (direct): each wave reconstructs CGRP/nwaves blocks
        for (int s = wavei; s < CGRP; s += nwaves) {
            long blk = base + s;
            if (blk >= nblocks) break;
            int bk = (int)(blk % b2);
            long t = blk / b2;
            int bj = (int)(t % b1);
            int bi = (int)(t / b1);
            int gi = bi * 4 + li;
            int gj = bj * 4 + lj;
            int gk = bk * 4 + lk;
            bool valid = (gi < d0) && (gj < d1) && (gk < d2);
            long idx = ((long)gi * d1 + gj) * d2 + gk;
            u32 hdr = s_hdr[s];
            float outv = 0.f;
            if (hdr >> 15) {
                int emax = (int)(hdr & 0x1FFu) - 256;
                u64 w = (lane < PLANES) ? s_planes[s][lane] : 0;
                u32 u = values_from_planes(w, lane);
                int q = (int)((u ^ NBMASK) - NBMASK);  // negabinary inv
                q = __shfl(q, ZIPERM[lane]);
                q = inv_axis(q, lane, 16);
                q = inv_axis(q, lane, 4);
                q = inv_axis(q, lane, 1);
                outv = ldexpf((float)q, emax - QBITS);
            }
            if (valid) {
                if (BF16_OUT)
                    ((bf16*)yv)[idx] = f2bf(outv);
                else
                    ((float*)yv)[idx] = outv;
            }
        }
        __syncthreads();               // LDS reused next round
    }
}

// ---------------------------------------------------------------------------
// fp8 (e4m3fn) wire codec: amax-scaled cast, 1 B/value + 4-byte scale.
// Wire layout (identical to the torch fallback in parallel/comm.py):
// bytes [0,4) = fp32 amax bits, bytes [4, 4+n) = e4m3 of x*448/amax.
// Three tiny kernels (init + atomic amax + cast) replace the torch
// path's ~5 dispatches and its extra fp32 materialization pass; the
// scale never visits the host.
// ---------------------------------------------------------------------------

__device__ __forceinline__ u8 f32_to_e4m3(float f) {
    u32 b = __float_as_uint(f);
    u8 sign = (u8)((b >> 24) & 0x80);
    u32 ab = b & 0x7fffffffu;
    if (ab > 0x7f800000u) return sign | 0x7f;          // NaN
    if (ab < 0x3C800000u) {                            // |x| < 2^-6
        // subnormal: m = rne(|x| * 2^9); m == 8 lands exactly on the
        // min-normal encoding, so the boundary needs no special case
        u32 m = (u32)rintf(__uint_as_float(ab) * 512.f);
        return sign | (u8)m;
    }
    // normal: RNE to 3 mantissa bits in the integer domain
    u32 r = ab + 0x0007FFFFu + ((ab >> 20) & 1u);
    u32 out = (r >> 20) - ((127u - 7u) << 3);
    if (out > 0x7e) out = 0x7e;                        // saturate to 448
    return sign | (u8)out;
}

__device__ __forceinline__ float e4m3_to_f32(u8 v) {
    u32 sign = (u32)(v & 0x80) << 24;
    u32 exp = (v >> 3) & 0xf;
    u32 man = v & 0x7;
    if (exp == 0xf && man == 0x7)
        return __uint_as_float(sign | 0x7fc00000u);    // NaN
    float m;
    if (exp == 0)
        m = (float)man * (1.f / 512.f);                // subnormal 2^-9
    else
        m = ldexpf(8.f + (float)man, (int)exp - 10);   // (1+man/8)*2^(e-7)
    return (v & 0x80) ? -m : m;
}

__global__ void fp8_init_kernel(float* amax) { *amax = 1e-12f; }

__global__ void fp8_amax_kernel(const bf16* __restrict__ x, long n,
                                float* __restrict__ amax) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    float m = 0.f;
    for (; i < n; i += (long)gridDim.x * blockDim.x) {
        float v = fabsf(bf2f(x[i]));
        if (isfinite(v)) m = fmaxf(m, v);
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        m = fmaxf(m, __shfl_xor(m, off));
    // LDS-reduce the block's waves to ONE atomic per block: per-wave
    // atomics to the single global word serialized at ~12 ns each —
    // 16K of them put a ~200 us floor under encode (measured)
    __shared__ float wmax[8];
    const int wv = threadIdx.x / WAVE;
    if ((threadIdx.x % WAVE) == 0) wmax[wv] = m;
    __syncthreads();
    if (threadIdx.x == 0) {
        const int nw = blockDim.x / WAVE;
        for (int w = 1; w < nw; ++w) m = fmaxf(m, wmax[w]);
        // non-negative floats compare correctly as uints
        atomicMax((unsigned*)amax, __float_as_uint(m));
    }
}

__global__ void fp8_cast_kernel(const bf16* __restrict__ x, long n,
                                const float* __restrict__ amax,
                                u8* __restrict__ out) {
    // correctly-rounded divide: fp32 '/' (and __fdiv_rn) lower to the
    // ~1-ulp v_rcp sequence here, which moved boundary values across
    // the e4m3 rounding point (measured: 4.9% byte mismatches vs the
    // torch wire). Device double division IS correctly rounded; the
    // once-per-tensor scale can afford it.
    float s = (float)(448.0 / (double)*amax);
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i < n; i += (long)gridDim.x * blockDim.x)
        out[i] = f32_to_e4m3(bf2f(x[i]) * s);
}

__global__ void fp8_decode_kernel(const u8* __restrict__ wire, long n,
                                  const float* __restrict__ amax,
                                  bf16* __restrict__ y) {
    float s = (float)((double)*amax / 448.0);
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i < n; i += (long)gridDim.x * blockDim.x)
        y[i] = f2bf(e4m3_to_f32(wire[i]) * s);
}

// ---------------------------------------------------------------------------
namespace defer_hip {

void launch_fp8_encode(const void* x, long n, void* out, hipStream_t s) {
    float* amax = (float*)out;                 // wire bytes [0, 4)
    u8* payload = (u8*)out + 4;
    hipLaunchKernelGGL(fp8_init_kernel, dim3(1), dim3(1), 0, s, amax);
    int grid = (int)((n + 255) / 256);
    if (grid > 2048) grid = 2048;   // one atomic per block after the
    hipLaunchKernelGGL(fp8_amax_kernel, dim3(grid), dim3(256), 0, s,
                       (const bf16*)x, n, amax);   // LDS reduce
    hipLaunchKernelGGL(fp8_cast_kernel, dim3(grid), dim3(256), 0, s,
                       (const bf16*)x, n, amax, payload);
}

void launch_fp8_decode(const void* wire, long n, void* y, hipStream_t s) {
    const float* amax = (const float*)wire;
    const u8* payload = (const u8*)wire + 4;
    int grid = (int)((n + 255) / 256);
    if (grid > 4096) grid = 4096;
    hipLaunchKernelGGL(fp8_decode_kernel, dim3(grid), dim3(256), 0, s,
                       payload, n, amax, (bf16*)y);
}

static int codec_grid(long nblocks, int blocks_per_wg) {
    long wgs = (nblocks + blocks_per_wg - 1) / blocks_per_wg;
    return (int)(wgs < 2048 ? (wgs > 0 ? wgs : 1) : 2048);
}

void launch_zfp_encode(const void* x, void* out, bool bf16_in, int d0,
                       int d1, int d2, int rate, hipStream_t s,
                       int phases) {
    int b0 = (d0 + 3) / 4, b1 = (d1 + 3) / 4, b2 = (d2 + 3) / 4;
    long nblocks = (long)b0 * b1 * b2;
    bool panel = (b2 % 16 == 0);      // NHWC boundary tensors: C >= 64
    dim3 grid(codec_grid(nblocks, 256)), block(256);
#define ENC_1(BF, PH, PAN)                                                 \
    hipLaunchKernelGGL((zfp_encode_kernel<BF, PH, PAN>), grid, block, 0,   \
                       s, x, (u32*)out, d0, d1, d2, b0, b1, b2, rate)
#define ENC_DISPATCH(BF)                                                   \
    do {                                                                   \
        if (phases == 1) {                                                 \
            if (panel) ENC_1(BF, 1, true); else ENC_1(BF, 1, false);       \
        } else if (phases == 4 && panel) {   /* bisect: stage+reduce */    \
            ENC_1(BF, 4, true);                                            \
        } else if (phases == 5 && panel) {   /* bisect: + lift */          \
            ENC_1(BF, 5, true);                                            \
        } else if (phases == 2) {                                          \
            if (panel) ENC_1(BF, 2, true); else ENC_1(BF, 2, false);       \
        } else {                                                           \
            if (panel) ENC_1(BF, 3, true); else ENC_1(BF, 3, false);       \
        }                                                                  \
    } while (0)
    if (bf16_in) ENC_DISPATCH(true);
    else ENC_DISPATCH(false);
#undef ENC_DISPATCH
#undef ENC_1
}

void launch_zfp_decode(const void* wire, void* y, bool bf16_out, int d0,
                       int d1, int d2, int rate, hipStream_t s) {
    int b0 = (d0 + 3) / 4, b1 = (d1 + 3) / 4, b2 = (d2 + 3) / 4;
    long nblocks = (long)b0 * b1 * b2;
    bool panel = (b2 % 16 == 0);
    dim3 grid(codec_grid(nblocks, 256)), block(256);
    if (bf16_out) {
        if (panel)
            hipLaunchKernelGGL((zfp_decode_kernel<true, true>), grid,
                               block, 0, s, (const u32*)wire, y, d0, d1,
                               d2, b0, b1, b2, rate);
        else
            hipLaunchKernelGGL((zfp_decode_kernel<true, false>), grid,
                               block, 0, s, (const u32*)wire, y, d0, d1,
                               d2, b0, b1, b2, rate);
    } else {
        if (panel)
            hipLaunchKernelGGL((zfp_decode_kernel<false, true>), grid,
                               block, 0, s, (const u32*)wire, y, d0, d1,
                               d2, b0, b1, b2, rate);
        else
            hipLaunchKernelGGL((zfp_decode_kernel<false, false>), grid,
                               block, 0, s, (const u32*)wire, y, d0, d1,
                               d2, b0, b1, b2, rate);
    }
}

}  // namespace defer_hip
