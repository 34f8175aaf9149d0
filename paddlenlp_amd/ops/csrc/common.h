// Common helpers for the paddlenlp_amd gfx950 (CDNA4) kernel pack.
//
// Conventions (see /opt/skills/guides/cdna_hip_programming.md):
//  - wave = 64 lanes, hard-coded
//  - bf16 loads always vectorized as short4/short8 (guide G13)
//  - block sizes are multiples of 64
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64

typedef __attribute__((ext_vector_type(2))) short short2v;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

typedef unsigned short ushort_t;

// ---- bf16 <-> fp32 ----
__device__ __forceinline__ float bf16_to_f32(ushort_t u) {
    union { unsigned int i; float f; } v;
    v.i = ((unsigned int)u) << 16;
    return v.f;
}

__device__ __forceinline__ ushort_t f32_to_bf16(float f) {
    // round-to-nearest-even via the HIP intrinsic type
    __hip_bfloat16 h = __float2bfloat16(f);
    union { __hip_bfloat16 h; ushort_t u; } v;
    v.h = h;
    return v.u;
}

// ---- wave-level reductions (64 lanes) ----
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, 64);
    return x;
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, 64));
    return x;
}

// block reduction via LDS, blockDim.x <= 1024 (<=16 waves)
__device__ __forceinline__ float block_reduce_sum(float x, float* lds_scratch) {
    int lane = threadIdx.x & 63;
    int wave = threadIdx.x >> 6;
    int nwaves = (blockDim.x + 63) >> 6;
    x = wave_reduce_sum(x);
    if (lane == 0) lds_scratch[wave] = x;
    __syncthreads();
    x = (threadIdx.x < nwaves) ? lds_scratch[threadIdx.x] : 0.0f;
    if (wave == 0) {
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) x += __shfl_xor(x, off, 64);
        if (lane == 0) lds_scratch[0] = x;
    }
    __syncthreads();
    float r = lds_scratch[0];
    __syncthreads();
    return r;
}

__device__ __forceinline__ float block_reduce_max(float x, float* lds_scratch) {
    int lane = threadIdx.x & 63;
    int wave = threadIdx.x >> 6;
    int nwaves = (blockDim.x + 63) >> 6;
    x = wave_reduce_max(x);
    if (lane == 0) lds_scratch[wave] = x;
    __syncthreads();
    x = (threadIdx.x < nwaves) ? lds_scratch[threadIdx.x] : -INFINITY;
    if (wave == 0) {
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, 64));
        if (lane == 0) lds_scratch[0] = x;
    }
    __syncthreads();
    float r = lds_scratch[0];
    __syncthreads();
    return r;
}

// grid sizing for memory-bound grid-stride kernels (guide G11):
// cap at 256 CU x 8 blocks and stride the rest.
static inline int memgrid(long long total_threads, int block) {
    long long blocks = (total_threads + block - 1) / block;
    long long cap = 2048;
    return (int)(blocks < cap ? blocks : cap);
}
