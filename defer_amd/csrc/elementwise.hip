// Elementwise NHWC kernels: batchnorm-apply, residual add+act, relu,
// row softmax. Memory-bound: bf16 loads/stores vectorized 8-wide
// (16 B/lane), grid-stride loops sized <= 2048 blocks.
//
// These replace the Keras BatchNormalization / Add / ReLU / softmax
// layers the reference runs inside model.predict
// (/root/reference/src/node.py:106).
#include "common.h"
#include "kernels.h"

// y = act(x * scale[c] + bias[c]); x [*, C] with C % 8 == 0
__global__ void bn_act_kernel(const bf16* __restrict__ x,
                              const float* __restrict__ scale,
                              const float* __restrict__ bias,
                              bf16* __restrict__ y,
                              long total8, int c8, int act) {
    long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = i0; i < total8; i += stride) {
        int c = (int)(i % c8) * 8;
        bf16x8 v = load_bf16x8(x + i * 8);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float s = scale ? scale[c + j] : 1.0f;
            float b = bias ? bias[c + j] : 0.0f;
            o[j] = f2bf(apply_act(bf2f(v[j]) * s + b, act));
        }
        store_bf16x8(y + i * 8, o);
    }
}

// y = act(a + b), both [total8*8] bf16
__global__ void add_act_kernel(const bf16* __restrict__ a,
                               const bf16* __restrict__ b,
                               bf16* __restrict__ y,
                               long total8, int act) {
    long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = i0; i < total8; i += stride) {
        bf16x8 va = load_bf16x8(a + i * 8);
        bf16x8 vb = load_bf16x8(b + i * 8);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j)
            o[j] = f2bf(apply_act(bf2f(va[j]) + bf2f(vb[j]), act));
        store_bf16x8(y + i * 8, o);
    }
}

__global__ void relu_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                            long total8) {
    long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = i0; i < total8; i += stride) {
        bf16x8 v = load_bf16x8(x + i * 8);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = f2bf(fmaxf(bf2f(v[j]), 0.0f));
        store_bf16x8(y + i * 8, o);
    }
}

// Row softmax: one wave per row, fp32 accumulation, arbitrary ncols.
__global__ void softmax_kernel(const bf16* __restrict__ x,
                               bf16* __restrict__ y, int rows, int cols) {
    int row = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
    int lane = threadIdx.x % WAVE;
    if (row >= rows) return;
    const bf16* xr = x + (long)row * cols;
    float m = -1e30f;
    for (int c = lane; c < cols; c += WAVE) m = fmaxf(m, bf2f(xr[c]));
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        m = fmaxf(m, __shfl_xor(m, off));
    float s = 0.f;
    for (int c = lane; c < cols; c += WAVE) s += __expf(bf2f(xr[c]) - m);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) s += __shfl_xor(s, off);
    float inv = 1.0f / s;
    bf16* yr = y + (long)row * cols;
    for (int c = lane; c < cols; c += WAVE)
        yr[c] = f2bf(__expf(bf2f(xr[c]) - m) * inv);
}


// Two-input channel concat, NHWC bf16 (DenseNet dense connections:
// cat = [prev, feat]). One bf16x8 chunk per thread; consecutive
// threads write consecutive output chunks, and each 8-chunk comes
// whole from one source (c1 % 8 == 0), so reads stay coalesced per
// segment — replaces per-input hipMemcpy2DAsync (small-pitch 2D
// copies measured 10.8% of the DenseNet step).
__global__ void cat2_kernel(const bf16* __restrict__ a,
                            const bf16* __restrict__ b,
                            bf16* __restrict__ y,
                            long rows, int c1_8, int c2_8) {
    const int ct8 = c1_8 + c2_8;
    long total = rows * ct8;
    long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long gstride = (long)gridDim.x * blockDim.x;
    for (long i = i0; i < total; i += gstride) {
        long r = i / ct8;
        int c8 = (int)(i % ct8);
        bf16x8 v = (c8 < c1_8)
            ? load_bf16x8(a + (r * c1_8 + c8) * 8)
            : load_bf16x8(b + (r * c2_8 + (c8 - c1_8)) * 8);
        store_bf16x8(y + i * 8, v);
    }
}

namespace defer_hip {

static int grid1d(long work, int block) {
    long g = (work + block - 1) / block;
    return (int)(g < 2048 ? g : 2048);
}

void launch_bn_act(const void* x, const float* scale, const float* bias,
                   void* y, long total8, int c8, bool relu, hipStream_t s) {
    hipLaunchKernelGGL(bn_act_kernel, dim3(grid1d(total8, 256)), dim3(256),
                       0, s, (const bf16*)x, scale, bias, (bf16*)y, total8,
                       c8, relu ? ACT_RELU : ACT_NONE);
}

void launch_add_act(const void* a, const void* b, void* y, long total8,
                    bool relu, hipStream_t s) {
    hipLaunchKernelGGL(add_act_kernel, dim3(grid1d(total8, 256)), dim3(256),
                       0, s, (const bf16*)a, (const bf16*)b, (bf16*)y,
                       total8, relu ? ACT_RELU : ACT_NONE);
}

void launch_relu(const void* x, void* y, long total8, hipStream_t s) {
    hipLaunchKernelGGL(relu_kernel, dim3(grid1d(total8, 256)), dim3(256), 0,
                       s, (const bf16*)x, (bf16*)y, total8);
}

void launch_cat2(const void* a, const void* b, void* y, long rows,
                 int c1, int c2, hipStream_t s) {
    long total = rows * ((c1 + c2) / 8);
    hipLaunchKernelGGL(cat2_kernel, dim3(grid1d(total, 256)), dim3(256),
                       0, s, (const bf16*)a, (const bf16*)b, (bf16*)y,
                       rows, c1 / 8, c2 / 8);
}

void launch_softmax(const void* x, void* y, int rows, int cols,
                    hipStream_t s) {
    int wpb = 4;
    int blocks = (rows + wpb - 1) / wpb;
    hipLaunchKernelGGL(softmax_kernel, dim3(blocks), dim3(wpb * 64), 0, s,
                       (const bf16*)x, (bf16*)y, rows, cols);
}

}  // namespace defer_hip
