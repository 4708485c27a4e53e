// Common device helpers for defer_amd gfx950 (CDNA4) kernels.
//
// All kernels in this library are written for MI355X only: wave64,
// MFMA bf16 (16x16x32), 160 KiB LDS per CU, NHWC activations.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

typedef __bf16 bf16;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x2 __attribute__((ext_vector_type(2)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef unsigned char u8;
typedef unsigned int u32;
typedef unsigned long long u64;
typedef short s16x8 __attribute__((ext_vector_type(8)));
typedef u32 u32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float bf2f(bf16 v) { return (float)v; }
__device__ __forceinline__ bf16 f2bf(float v) { return (bf16)v; }

// Load 8 bf16 (16 B) as one vector
__device__ __forceinline__ bf16x8 load_bf16x8(const bf16* p) {
    return *reinterpret_cast<const bf16x8*>(p);
}
__device__ __forceinline__ void store_bf16x8(bf16* p, bf16x8 v) {
    *reinterpret_cast<bf16x8*>(p) = v;
}

__device__ __forceinline__ int cdiv(int a, int b) { return (a + b - 1) / b; }

#define MFMA_BF16_16x16x32(a, b, c) \
    __builtin_amdgcn_mfma_f32_16x16x32_bf16((a), (b), (c), 0, 0, 0)

// activation codes
enum { ACT_NONE = 0, ACT_RELU = 1 };

__device__ __forceinline__ float apply_act(float v, int act) {
    return act == ACT_RELU ? fmaxf(v, 0.0f) : v;
}
