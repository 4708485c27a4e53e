// Pooling kernels, NHWC bf16: max pool (ResNet50 stem 3x3/2, VGG 2x2/2)
// and global average pool (LDS-free: channels are the fast dim, so each
// lane reduces its own 8 channels over H*W — coalesced and conflict-free).
//
// Replaces Keras MaxPooling2D / GlobalAveragePooling2D
// (/root/reference/test/test.py:14 ResNet50 stem/head).
#include "common.h"
#include "kernels.h"

// out[n][oh][ow][c] = max over window. C % 8 == 0; one lane handles 8
// channels of one output pixel.
__global__ void maxpool_kernel(const bf16* __restrict__ x,
                               bf16* __restrict__ y,
                               int N, int H, int W, int C,
                               int OH, int OW, int kh, int kw,
                               int stride, int pad) {
    long total = (long)N * OH * OW * (C / 8);
    long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long gstride = (long)gridDim.x * blockDim.x;
    for (long i = i0; i < total; i += gstride) {
        int c8 = (int)(i % (C / 8));
        long p = i / (C / 8);
        int ow = (int)(p % OW);
        long q = p / OW;
        int oh = (int)(q % OH);
        int n = (int)(q / OH);
        float best[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) best[j] = -1e30f;
        int ih0 = oh * stride - pad, iw0 = ow * stride - pad;
        for (int r = 0; r < kh; ++r) {
            int ih = ih0 + r;
            if (ih < 0 || ih >= H) continue;
            for (int s = 0; s < kw; ++s) {
                int iw = iw0 + s;
                if (iw < 0 || iw >= W) continue;
                bf16x8 v = load_bf16x8(
                    x + (((long)n * H + ih) * W + iw) * C + c8 * 8);
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    best[j] = fmaxf(best[j], bf2f(v[j]));
            }
        }
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = f2bf(best[j]);
        store_bf16x8(y + (((long)n * OH + oh) * OW + ow) * C + c8 * 8, o);
    }
}

// out[n][oh][ow][c] = mean over window (valid taps only — matches
// count_include_pad=False; DenseNet transition 2x2/2 uses pad 0 where
// the distinction vanishes). Same lane mapping as maxpool.
__global__ void avgpool_kernel(const bf16* __restrict__ x,
                               bf16* __restrict__ y,
                               int N, int H, int W, int C,
                               int OH, int OW, int kh, int kw,
                               int stride, int pad) {
    long total = (long)N * OH * OW * (C / 8);
    long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long gstride = (long)gridDim.x * blockDim.x;
    for (long i = i0; i < total; i += gstride) {
        int c8 = (int)(i % (C / 8));
        long p = i / (C / 8);
        int ow = (int)(p % OW);
        long q = p / OW;
        int oh = (int)(q % OH);
        int n = (int)(q / OH);
        float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
        int cnt = 0;
        int ih0 = oh * stride - pad, iw0 = ow * stride - pad;
        for (int r = 0; r < kh; ++r) {
            int ih = ih0 + r;
            if (ih < 0 || ih >= H) continue;
            for (int s = 0; s < kw; ++s) {
                int iw = iw0 + s;
                if (iw < 0 || iw >= W) continue;
                bf16x8 v = load_bf16x8(
                    x + (((long)n * H + ih) * W + iw) * C + c8 * 8);
#pragma unroll
                for (int j = 0; j < 8; ++j) acc[j] += bf2f(v[j]);
                ++cnt;
            }
        }
        float inv = cnt > 0 ? 1.0f / cnt : 0.f;
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = f2bf(acc[j] * inv);
        store_bf16x8(y + (((long)n * OH + oh) * OW + ow) * C + c8 * 8, o);
    }
}

// [N, H, W, C] -> [N, C]; one workgroup per (n, c-block of 2048),
// each lane averages its own 8 channels over all H*W pixels.
__global__ void gap_kernel(const bf16* __restrict__ x,
                           bf16* __restrict__ y,
                           int N, int HW, int C) {
    int cblk = (C + 2047) / 2048;
    int n = blockIdx.x / cblk;
    int cb = blockIdx.x % cblk;
    int c = cb * 2048 + threadIdx.x * 8;
    if (n >= N || c >= C) return;
    const bf16* xn = x + (long)n * HW * C + c;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int p = 0; p < HW; ++p) {
        bf16x8 v = load_bf16x8(xn + (long)p * C);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += bf2f(v[j]);
    }
    float inv = 1.0f / HW;
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(acc[j] * inv);
    store_bf16x8(y + (long)n * C + c, o);
}

namespace defer_hip {

static int grid1d(long work, int block) {
    long g = (work + block - 1) / block;
    return (int)(g < 2048 ? g : 2048);
}

void launch_maxpool(const void* x, void* y, int NB, int H, int W, int C,
                    int OH, int OW, int k, int stride, int pad,
                    hipStream_t s) {
    long total = (long)NB * OH * OW * (C / 8);
    hipLaunchKernelGGL(maxpool_kernel, dim3(grid1d(total, 256)), dim3(256),
                       0, s, (const bf16*)x, (bf16*)y, NB, H, W, C, OH, OW,
                       k, k, stride, pad);
}

void launch_avgpool(const void* x, void* y, int NB, int H, int W, int C,
                    int OH, int OW, int k, int stride, int pad,
                    hipStream_t s) {
    long total = (long)NB * OH * OW * (C / 8);
    hipLaunchKernelGGL(avgpool_kernel, dim3(grid1d(total, 256)),
                       dim3(256), 0, s, (const bf16*)x, (bf16*)y, NB, H,
                       W, C, OH, OW, k, k, stride, pad);
}

void launch_gap(const void* x, void* y, int NB, int HW, int C,
                hipStream_t s) {
    int cblk = (C + 2047) / 2048;
    hipLaunchKernelGGL(gap_kernel, dim3(NB * cblk), dim3(256), 0, s,
                       (const bf16*)x, (bf16*)y, NB, HW, C);
}

}  // namespace defer_hip
