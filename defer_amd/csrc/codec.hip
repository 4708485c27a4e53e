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
template <bool BF16_IN>
__global__ void zfp_encode_kernel(const void* __restrict__ xv,
                                  u32* __restrict__ out, int d0, int d1,
                                  int d2, int b0, int b1, int b2,
                                  int rate) {
    const int lane = threadIdx.x % WAVE;
    const int wavei = threadIdx.x / WAVE;
    const int nwaves = blockDim.x / WAVE;
    const long nblocks = (long)b0 * b1 * b2;
    const int li = lane >> 4, lj = (lane >> 2) & 3, lk = lane & 3;
    const int wpb = rate * 2;                   // u32 words per block

    for (long blk = (long)blockIdx.x * nwaves + wavei; blk < nblocks;
         blk += (long)gridDim.x * nwaves) {
        int bk = (int)(blk % b2);
        long t = blk / b2;
        int bj = (int)(t % b1);
        int bi = (int)(t / b1);
        int gi = bi * 4 + li; if (gi > d0 - 1) gi = d0 - 1;
        int gj = bj * 4 + lj; if (gj > d1 - 1) gj = d1 - 1;
        int gk = bk * 4 + lk; if (gk > d2 - 1) gk = d2 - 1;
        long idx = ((long)gi * d1 + gj) * d2 + gk;
        float v = BF16_IN ? bf2f(((const bf16*)xv)[idx])
                          : ((const float*)xv)[idx];

        // max |v| over the wave
        float av = fabsf(v);
#pragma unroll
        for (int off = 32; off > 0; off >>= 1)
            av = fmaxf(av, __shfl_xor(av, off));

        u32* bout = out + blk * wpb;
        if (!(av > 0.f) || !isfinite(av)) {     // zero/non-finite block
            if (lane == 0)
                for (int w = 0; w < wpb; ++w) bout[w] = 0;
            continue;
        }
        int emax;
        frexpf(av, &emax);
        // quantize (f32 product, round-half-even — matches numpy rint)
        int q = (int)rintf(v * ldexpf(1.0f, QBITS - emax));
        // transform along k (stride 1), j (4), i (16)
        q = fwd_axis(q, lane, 1);
        q = fwd_axis(q, lane, 4);
        q = fwd_axis(q, lane, 16);
        // sequency reorder: stream slot `lane` holds coef ZPERM[lane]
        q = __shfl(q, ZPERM[lane]);
        // negabinary
        u32 u = ((u32)q + NBMASK) ^ NBMASK;

        BitWriter wr{bout, 0u, 0, rate * 64};
        if (lane == 0)
            wr.put_bits((1u << 15) | ((u32)(emax + 256) & 0x1FFu),
                        HDR_BITS);
        int n = 0;
        for (int p = PLANES - 1; p >= 0 && wr.budget > 0; --p) {
            u64 x = __ballot((u >> p) & 1);
            if (lane == 0) {
                // significant prefix
                wr.put_bits(x & ((n < 64) ? ((1ull << n) - 1ull)
                                          : ~0ull), n);
                x >>= n;
                // group tests, run-wise: each run is tz zeros then a
                // one = the word (1 << tz) over tz+1 bits (bit-identical
                // to per-bit emission; put_bits truncates at the budget
                // exactly like per-bit puts would)
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
            // all lanes need n for the next plane's ballot bookkeeping?
            // n only lives on lane 0; broadcast not needed (only lane 0
            // uses it), but the loop break on budget must be uniform:
            n = __shfl(n, 0);
            int bud = __shfl(wr.budget, 0);
            if (lane != 0) wr.budget = bud;
        }
        if (lane == 0) wr.finish(bout + wpb);
    }
}

template <bool BF16_OUT>
__global__ void zfp_decode_kernel(const u32* __restrict__ wire,
                                  void* __restrict__ yv, int d0, int d1,
                                  int d2, int b0, int b1, int b2,
                                  int rate) {
    const int lane = threadIdx.x % WAVE;
    const int wavei = threadIdx.x / WAVE;
    const int nwaves = blockDim.x / WAVE;
    const long nblocks = (long)b0 * b1 * b2;
    const int li = lane >> 4, lj = (lane >> 2) & 3, lk = lane & 3;
    const int wpb = rate * 2;

    for (long blk = (long)blockIdx.x * nwaves + wavei; blk < nblocks;
         blk += (long)gridDim.x * nwaves) {
        int bk = (int)(blk % b2);
        long t = blk / b2;
        int bj = (int)(t % b1);
        int bi = (int)(t / b1);
        int gi = bi * 4 + li;
        int gj = bj * 4 + lj;
        int gk = bk * 4 + lk;
        bool valid = (gi < d0) && (gj < d1) && (gk < d2);
        long idx = ((long)gi * d1 + gj) * d2 + gk;

        const u32* bin = wire + blk * wpb;
        // lane 0 parses word-wise; plane words broadcast to all lanes.
        // Truncation endgame differs from the per-bit reference only in
        // zero bits / internal n, so reconstruction is identical.
        BitReader rd;
        rd.init(bin, rate * 64);
        u32 hdr = 0;
        if (lane == 0) hdr = (u32)rd.take(HDR_BITS);
        hdr = __shfl(hdr, 0);
        float outv = 0.f;
        if (hdr >> 15) {
            int emax = (int)(hdr & 0x1FFu) - 256;
            u32 u = 0;
            int n = 0;
            int done = 0;
            for (int p = PLANES - 1; p >= 0; --p) {
                u64 x = 0;
                if (lane == 0) {
                    if (rd.empty()) {
                        done = 1;
                    } else {
                        x = rd.take(n);               // significant prefix
                        while (n < 64) {              // group runs
                            if (rd.empty()) break;
                            if (!rd.take(1)) break;   // group test
                            // scan zeros (may span window refills) until
                            // the run's 1 bit
                            bool found = false;
                            while (n < 64) {
                                rd.refill();
                                int lim = 64 - n;
                                int avail =
                                    rd.nwin < lim ? rd.nwin : lim;
                                if (avail == 0) break;  // stream end
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
                                rd.take(avail);       // all zeros
                                n += avail;
                                if (rd.empty()) break;
                            }
                            if (!found) break;
                        }
                    }
                }
                done = __shfl(done, 0);
                if (done) break;
                x = __shfl(x, 0);
                n = __shfl(n, 0);
                u |= (u32)((x >> lane) & 1) << p;
            }
            int q = (int)((u ^ NBMASK) - NBMASK);   // negabinary inverse
            // inverse sequency: coef position `lane` from stream slot
            // ZIPERM[lane]
            q = __shfl(q, ZIPERM[lane]);
            // inverse transform: axes i, j, k
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
}

// ---------------------------------------------------------------------------
namespace defer_hip {

static int codec_grid(long nblocks, int waves_per_wg) {
    long wgs = (nblocks + waves_per_wg - 1) / waves_per_wg;
    return (int)(wgs < 2048 ? (wgs > 0 ? wgs : 1) : 2048);
}

void launch_zfp_encode(const void* x, void* out, bool bf16_in, int d0,
                       int d1, int d2, int rate, hipStream_t s) {
    int b0 = (d0 + 3) / 4, b1 = (d1 + 3) / 4, b2 = (d2 + 3) / 4;
    long nblocks = (long)b0 * b1 * b2;
    dim3 grid(codec_grid(nblocks, 4)), block(256);
    if (bf16_in)
        hipLaunchKernelGGL((zfp_encode_kernel<true>), grid, block, 0, s,
                           x, (u32*)out, d0, d1, d2, b0, b1, b2, rate);
    else
        hipLaunchKernelGGL((zfp_encode_kernel<false>), grid, block, 0, s,
                           x, (u32*)out, d0, d1, d2, b0, b1, b2, rate);
}

void launch_zfp_decode(const void* wire, void* y, bool bf16_out, int d0,
                       int d1, int d2, int rate, hipStream_t s) {
    int b0 = (d0 + 3) / 4, b1 = (d1 + 3) / 4, b2 = (d2 + 3) / 4;
    long nblocks = (long)b0 * b1 * b2;
    dim3 grid(codec_grid(nblocks, 4)), block(256);
    if (bf16_out)
        hipLaunchKernelGGL((zfp_decode_kernel<true>), grid, block, 0, s,
                           (const u32*)wire, y, d0, d1, d2, b0, b1, b2,
                           rate);
    else
        hipLaunchKernelGGL((zfp_decode_kernel<false>), grid, block, 0, s,
                           (const u32*)wire, y, d0, d1, d2, b0, b1, b2,
                           rate);
}

}  // namespace defer_hip
