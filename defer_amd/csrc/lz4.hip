// LZ4-style byte compressor for gfx950 — the MI355X-native rebuild of the
// reference's lz4.frame stage over the ZFP payload
// (/root/reference/src/dispatcher.py:81-84, node.py:107: lz4(zfp(x))).
//
// Format (block-independent so compression is embarrassingly parallel,
// bit-exact to the Python spec in defer_amd/ops/lz4_ref.py):
//   header: u32 raw_len | u32 nblocks | u32 off[nblocks+1]
//   body:   concatenated per-block LZ4 sequences, block i at off[i]
// Each 4096-byte input block compresses independently (no cross-block
// matches). A sequence is the standard LZ4 block-format shape: token byte
// (hi nibble literal count, lo nibble match length - 4, 15 = extension
// bytes of 255), literals, u16 LE offset, match extension bytes. The final
// sequence of a block is literals-only; the decoder stops when the block's
// raw size is reached.
//
// Kernel shape: ONE 64-lane wavefront per block (workgroup = 1 wave so
// __syncthreads() is a wave-cheap barrier). The greedy hash-chain parse is
// inherently sequential -> lane 0 drives it over an LDS copy of the block;
// literal emission and block gather/scatter are wave-parallel. Blocks give
// the cross-CU parallelism: a 100 MB activation is ~25K blocks.
#include "common.h"
#include "kernels.h"

#define LZ_BLK 4096
#define LZ_MAXC 4352            // worst case 4096 + 4096/255 + 16, padded
#define LZ_HBITS 11
#define LZ_HSIZE (1 << LZ_HBITS)
#define LZ_HDR(nblocks) (4 * (2 + (nblocks) + 1))

typedef unsigned char u8;
typedef unsigned short u16;

__device__ __forceinline__ u32 lz_hash(u32 v) {
    return (v * 2654435761u) >> (32 - LZ_HBITS);
}

__device__ __forceinline__ u32 lds_read32(const u8* b, int p) {
    return (u32)b[p] | ((u32)b[p + 1] << 8) | ((u32)b[p + 2] << 16) |
           ((u32)b[p + 3] << 24);
}

// ---- compress: one wave per LZ_BLK block --------------------------------
__global__ void __launch_bounds__(64)
lz4_compress_kernel(const u8* __restrict__ in, long n,
                    u8* __restrict__ scratch, u32* __restrict__ sizes) {
    __shared__ u8 buf[LZ_BLK];
    __shared__ u16 tab[LZ_HSIZE];
    const int lane = threadIdx.x;
    const long base = (long)blockIdx.x * LZ_BLK;
    const int len = (int)((n - base) < LZ_BLK ? (n - base) : LZ_BLK);
    u8* out = scratch + (long)blockIdx.x * LZ_MAXC;

    for (int i = lane; i < len; i += WAVE) buf[i] = in[base + i];
    for (int i = lane; i < LZ_HSIZE; i += WAVE) tab[i] = 0xFFFF;
    __syncthreads();

    int opos = 0;     // kept uniform across the wave via shfl
    int pos = 0, anchor = 0;
    const int mend = len - 5;   // matches may not extend into last 5 bytes

    for (;;) {
        // lane 0 scans for the next match (or end of block)
        int litFrom = 0, litLen = 0, moff = 0, mlen = 0, done = 0;
        if (lane == 0) {
            // skip acceleration (mirrors ops/lz4_ref.py): after every 64
            // failed probes the scan step grows by one
            int cnt = 64;
            for (;;) {
                if (pos >= len - 8) {  // tail: emit final literals
                    litFrom = anchor; litLen = len - anchor;
                    done = 1;
                    break;
                }
                u32 v = lds_read32(buf, pos);
                u32 h = lz_hash(v);
                int cand = tab[h];
                tab[h] = (u16)pos;
                if (cand != 0xFFFF && lds_read32(buf, cand) == v) {
                    mlen = 4;
                    while (pos + mlen < mend &&
                           buf[cand + mlen] == buf[pos + mlen])
                        ++mlen;
                    litFrom = anchor; litLen = pos - anchor;
                    moff = pos - cand;
                    pos += mlen;
                    anchor = pos;
                    break;
                }
                pos += cnt >> 6;
                ++cnt;
            }
        }
        litFrom = __shfl(litFrom, 0);
        litLen = __shfl(litLen, 0);
        moff = __shfl(moff, 0);
        mlen = __shfl(mlen, 0);
        done = __shfl(done, 0);

        // ---- emit sequence: token + ext litlen (lane 0)
        int mtok = done ? 0 : (mlen - 4);
        int hdr = opos;
        int nlit_ext = (litLen >= 15) ? (litLen - 15) / 255 + 1 : 0;
        if (lane == 0) {
            out[hdr] = (u8)(((litLen < 15 ? litLen : 15) << 4) |
                            (mtok < 15 ? mtok : 15));
            int rem = litLen - 15;
            for (int i = 1; i <= nlit_ext; ++i) {
                out[hdr + i] = (u8)(rem < 255 ? rem : 255);
                rem -= 255;
            }
        }
        int lit0 = hdr + 1 + nlit_ext;
        // ---- literals: wave-parallel copy LDS -> global
        for (int i = lane; i < litLen; i += WAVE)
            out[lit0 + i] = buf[litFrom + i];
        opos = lit0 + litLen;
        if (done) break;
        // ---- offset + match length extension (lane 0)
        int next = opos + 2 + ((mtok >= 15) ? (mtok - 15) / 255 + 1 : 0);
        if (lane == 0) {
            out[opos] = (u8)(moff & 0xFF);
            out[opos + 1] = (u8)(moff >> 8);
            int rem = mtok - 15;
            int i = opos + 2;
            while (rem >= 0) {
                out[i++] = (u8)(rem < 255 ? rem : 255);
                rem -= 255;
            }
        }
        opos = next;
    }
    if (lane == 0) sizes[blockIdx.x] = (u32)opos;
}

// ---- offsets: one workgroup exclusive-scans the block sizes -------------
__global__ void __launch_bounds__(256)
lz4_offsets_kernel(const u32* __restrict__ sizes, u32* __restrict__ header,
                   int nblocks, u32 raw_len) {
    // header = [raw_len, nblocks, off[0..nblocks]]
    __shared__ u32 carry;
    __shared__ u32 ch[256];
    if (threadIdx.x == 0) {
        header[0] = raw_len;
        header[1] = (u32)nblocks;
        carry = 0;
    }
    __syncthreads();
    for (int b0 = 0; b0 < nblocks; b0 += 256) {
        int i = b0 + threadIdx.x;
        u32 v = (i < nblocks) ? sizes[i] : 0;
        // inclusive scan over the 256-chunk in LDS
        ch[threadIdx.x] = v;
        __syncthreads();
        for (int d = 1; d < 256; d <<= 1) {
            u32 add = (threadIdx.x >= d) ? ch[threadIdx.x - d] : 0;
            __syncthreads();
            ch[threadIdx.x] += add;
            __syncthreads();
        }
        if (i < nblocks)
            header[2 + i] = carry + ch[threadIdx.x] - v;   // exclusive
        if (i == nblocks - 1)
            header[2 + nblocks] = carry + ch[threadIdx.x]; // total
        __syncthreads();
        if (threadIdx.x == 0) carry += ch[255];
        __syncthreads();
    }
    if (nblocks == 0 && threadIdx.x == 0) header[2] = 0;
}

// ---- gather: compact per-block streams behind the header ----------------
__global__ void __launch_bounds__(64)
lz4_gather_kernel(const u8* __restrict__ scratch,
                  const u32* __restrict__ header, u8* __restrict__ out,
                  int nblocks) {
    const int lane = threadIdx.x;
    for (int b = blockIdx.x; b < nblocks; b += gridDim.x) {
        u32 off = header[2 + b], end = header[2 + b + 1];
        const u8* src = scratch + (long)b * LZ_MAXC;
        u8* dst = out + LZ_HDR(nblocks) + off;
        for (u32 i = lane; i < end - off; i += WAVE) dst[i] = src[i];
    }
}

// ---- decompress: one wave per block -------------------------------------
__global__ void __launch_bounds__(64)
lz4_decompress_kernel(const u8* __restrict__ comp, u8* __restrict__ out,
                      long raw_len) {
    __shared__ u8 dst[LZ_BLK];
    const int lane = threadIdx.x;
    const u32* header = (const u32*)comp;
    const int nblocks = (int)header[1];
    const int b = blockIdx.x;
    if (b >= nblocks) return;
    const long base = (long)b * LZ_BLK;
    const int blen = (int)((raw_len - base) < LZ_BLK ? (raw_len - base)
                                                     : LZ_BLK);
    const u8* src = comp + LZ_HDR(nblocks) + header[2 + b];
    const int clen = (int)(header[2 + b + 1] - header[2 + b]);

    int spos = 0, dpos = 0;
    while (dpos < blen && spos < clen) {
        // token + literal length (lane 0 parses, broadcast)
        int litLen = 0, mlen = 0, moff = 0, ns = 0;
        if (lane == 0) {
            int p = spos;
            u8 tok = src[p++];
            litLen = tok >> 4;
            if (litLen == 15) {
                u8 e;
                do { e = src[p++]; litLen += e; } while (e == 255);
            }
            ns = p;
            mlen = tok & 0xF;
        }
        litLen = __shfl(litLen, 0);
        ns = __shfl(ns, 0);
        // literals: wave-parallel copy global comp -> LDS
        for (int i = lane; i < litLen; i += WAVE)
            dst[dpos + i] = src[ns + i];
        __syncthreads();
        dpos += litLen;
        spos = ns + litLen;
        if (dpos >= blen) break;        // final literals-only sequence
        if (lane == 0) {
            int p = spos;
            moff = src[p] | ((int)src[p + 1] << 8);
            p += 2;
            int ml = mlen + 4;
            if (mlen == 15) {
                u8 e;
                do { e = src[p++]; ml += e; } while (e == 255);
            }
            mlen = ml;
            ns = p;
            // match copy: sequential byte copy handles overlap exactly
            // (LZ77 semantics: offset < mlen repeats the window)
            for (int i = 0; i < mlen; ++i)
                dst[dpos + i] = dst[dpos - moff + i];
        }
        mlen = __shfl(mlen, 0);
        ns = __shfl(ns, 0);
        __syncthreads();
        dpos += mlen;
        spos = ns;
    }
    __syncthreads();
    for (int i = lane; i < blen; i += WAVE) out[base + i] = dst[i];
}

// ---- wire length: total stream bytes as a device-side int64 -------------
// Lets the hop ship the size message straight from device memory (RCCL
// send of a device tensor) with no host round-trip on the critical path.
__global__ void
lz4_wire_len_kernel(const u32* __restrict__ header, int nblocks,
                    long long* __restrict__ len_out) {
    if (threadIdx.x == 0)
        *len_out = (long long)LZ_HDR(nblocks) +
                   (long long)header[2 + nblocks];
}

// ---------------------------------------------------------------------------
namespace defer_hip {

long lz4_max_compressed(long n) {
    long nb = (n + LZ_BLK - 1) / LZ_BLK;
    return LZ_HDR(nb) + nb * (long)LZ_MAXC;
}

long lz4_scratch_bytes(long n) {
    long nb = (n + LZ_BLK - 1) / LZ_BLK;
    return nb * (long)LZ_MAXC + nb * 4 /* sizes */;
}

void launch_lz4_compress(const void* in, long n, void* scratch, void* out,
                         hipStream_t s) {
    int nb = (int)((n + LZ_BLK - 1) / LZ_BLK);
    if (nb <= 0) return;
    u8* scr = (u8*)scratch;
    u32* sizes = (u32*)(scr + (long)nb * LZ_MAXC);
    hipLaunchKernelGGL(lz4_compress_kernel, dim3(nb), dim3(64), 0, s,
                       (const u8*)in, n, scr, sizes);
    hipLaunchKernelGGL(lz4_offsets_kernel, dim3(1), dim3(256), 0, s,
                       sizes, (u32*)out, nb, (u32)n);
    int gather_grid = nb < 4096 ? nb : 4096;
    hipLaunchKernelGGL(lz4_gather_kernel, dim3(gather_grid), dim3(64), 0,
                       s, scr, (const u32*)out, (u8*)out, nb);
}

void launch_lz4_wire_len(const void* out_stream, long n, void* len_out,
                         hipStream_t s) {
    int nb = (int)((n + LZ_BLK - 1) / LZ_BLK);
    hipLaunchKernelGGL(lz4_wire_len_kernel, dim3(1), dim3(1), 0, s,
                       (const u32*)out_stream, nb, (long long*)len_out);
}

void launch_lz4_decompress(const void* comp, void* out, long raw_len,
                           hipStream_t s) {
    int nb = (int)((raw_len + LZ_BLK - 1) / LZ_BLK);
    if (nb <= 0) return;
    hipLaunchKernelGGL(lz4_decompress_kernel, dim3(nb), dim3(64), 0, s,
                       (const u8*)comp, (u8*)out, raw_len);
}

}  // namespace defer_hip
