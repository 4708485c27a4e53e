// Host-visible launcher API for the defer_amd gfx950 kernel library.
// Implemented in the .hip TUs; called from the torch bindings.
#pragma once

#include <hip/hip_runtime.h>

namespace defer_hip {

// bf16 pointers are passed as void* across the host boundary.
struct ConvParams {
    const void* x;      // [NB, H, W, Cin] bf16 (or [M][K] in gemm mode)
    const void* w;      // [Cout][K] bf16 (OHWI flattened)
    const float* scale; // [Cout] or null
    const float* bias;  // [Cout] or null
    const void* res;    // [M][Cout] bf16 or null
    void* out;          // [M][Cout] bf16
    const void* zbuf;   // >=16 B zeros
    int M, K, Cout;
    int NB, H, W, Cin;
    int OH, OW, R, S, stride, pad;
    // magic-multiply reciprocals (filled by launch_conv_igemm):
    // floor(n/d) = umulhi(n, ceil(2^32/d)) for n*d < 2^32.
    // The window kernels do not divide; they reuse `smul` as a
    // full-sync debug flag (DEFER_CONV_VARIANT=W forces vmcnt(0)
    // drains every phase).
    unsigned int owmul, ohmul, cmul, smul;
};

void launch_conv_igemm(const ConvParams& p, bool relu, bool has_res,
                       bool gemm_mode, bool stem_mode, hipStream_t s);
void launch_pad_channels(const void* x, void* y, long rows, int C, int C8,
                         hipStream_t s);
// STEM path helpers: spatial zero-pad + weight repack to (r, run) K-order
void launch_pad2d(const void* x, void* y, int NB, int H, int W, int C,
                  int PH, int PW, int ph0, int pw0, hipStream_t s);
// small-Cin window stem path (R in {3,7}, stride in {1,2}, Cout 64,
// Cin pre-padded to 8; weights pre-repacked j-major zero-padded).
// Returns false if the shape does not qualify (caller falls back).
bool launch_conv_swin(const ConvParams& p, bool relu, int R, int stride,
                      hipStream_t s);
void launch_swin_repack_w(const void* w, void* wp, int Cout, int R,
                          int C, int Kpad, hipStream_t s);
void launch_stem_repack_w(const void* w, void* wp, int Cout, int R, int S,
                          int C, int TR, hipStream_t s);

// fused pre-activation 1x1 conv: out = relu(x*ps+pb) @ w^T
// (DenseNet BNActConv; bit-identical to bn_act -> conv GEMM mode)
void launch_gemm_prebn(const void* x, const void* w, const float* ps,
                       const float* pb, const float* os, const float* ob,
                       const void* zbuf, void* out, int M, int K,
                       int Cout, hipStream_t s);

void launch_bn_act(const void* x, const float* scale, const float* bias,
                   void* y, long total8, int c8, bool relu, hipStream_t s);
void launch_add_act(const void* a, const void* b, void* y, long total8,
                    bool relu, hipStream_t s);
void launch_relu(const void* x, void* y, long total8, hipStream_t s);
// two-input NHWC channel concat (c1, c2 % 8 == 0)
void launch_cat2(const void* a, const void* b, void* y, long rows,
                 int c1, int c2, hipStream_t s);
void launch_softmax(const void* x, void* y, int rows, int cols,
                    hipStream_t s);
void launch_maxpool(const void* x, void* y, int NB, int H, int W, int C,
                    int OH, int OW, int k, int stride, int pad,
                    hipStream_t s);
void launch_avgpool(const void* x, void* y, int NB, int H, int W, int C,
                    int OH, int OW, int k, int stride, int pad,
                    hipStream_t s);
void launch_gap(const void* x, void* y, int NB, int HW, int C,
                hipStream_t s);

// fixed-rate ZFP-style codec (see csrc/codec.hip / ops/zfp_ref.py)
// phases: 3 = full encode; 1/2 run only the transform / serialize phase
// (perf-bisection harness for tools/codecbench.py --phases)
void launch_zfp_encode(const void* x, void* out, bool bf16_in, int d0,
                       int d1, int d2, int rate, hipStream_t s,
                       int phases = 3);
void launch_zfp_decode(const void* wire, void* y, bool bf16_out, int d0,
                       int d1, int d2, int rate, hipStream_t s);

// fp8 e4m3fn wire codec: [4-byte fp32 amax][n bytes e4m3 of x*448/amax]
// (bit-compatible with the torch fallback in parallel/comm.py)
void launch_fp8_encode(const void* x, long n, void* out, hipStream_t s);
void launch_fp8_decode(const void* wire, long n, void* y, hipStream_t s);

// LZ4-style block compressor (see csrc/lz4.hip / ops/lz4_ref.py)
long lz4_max_compressed(long n);
long lz4_scratch_bytes(long n);
void launch_lz4_compress(const void* in, long n, void* scratch, void* out,
                         hipStream_t s);
// writes the total wire length (header + payload bytes) of a stream
// produced by launch_lz4_compress into *len_out (device int64), async
void launch_lz4_wire_len(const void* out_stream, long n, void* len_out,
                         hipStream_t s);
void launch_lz4_decompress(const void* comp, void* out, long raw_len,
                           hipStream_t s);

}  // namespace defer_hip
