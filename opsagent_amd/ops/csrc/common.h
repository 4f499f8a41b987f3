// Common helpers for opsagent_amd HIP kernels — MI355X (gfx950, CDNA4) only.
//
// Design per /opt/skills/guides/cdna_hip_programming.md:
//  * wave = 64 lanes; block sizes are multiples of 64
//  * bf16 loads ALWAYS vectorized as ushort4/ushort8 (guide G13: scalar bf16
//    loads cost ~2-2.5x on memory-bound kernels)
//  * cross-lane reduction via __shfl_xor over the 64-wide wave
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE 64

typedef __hip_bfloat16 bf16_t;

// 16-byte vector of 8 bf16 values (one ds_read_b128 / global dwordx4 worth)
struct alignas(16) bf16x8 {
    uint32_t u[4];
};
struct alignas(8) bf16x4 {
    uint32_t u[2];
};

__device__ __forceinline__ float bf16_to_f32(uint16_t b) {
    union {
        uint32_t u;
        float f;
    } v;
    v.u = ((uint32_t)b) << 16;
    return v.f;
}

__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
    union {
        uint32_t u;
        float f;
    } v;
    v.f = f;
    // round-to-nearest-even
    uint32_t lsb = (v.u >> 16) & 1u;
    v.u += 0x7fffu + lsb;
    return (uint16_t)(v.u >> 16);
}

// unpack a 32-bit word holding two bf16 (lo = element 0)
__device__ __forceinline__ float bf16_lo(uint32_t w) { return bf16_to_f32((uint16_t)(w & 0xffffu)); }
__device__ __forceinline__ float bf16_hi(uint32_t w) { return bf16_to_f32((uint16_t)(w >> 16)); }

__device__ __forceinline__ uint32_t pack_bf16x2(float lo, float hi) {
    return (uint32_t)f32_to_bf16(lo) | ((uint32_t)f32_to_bf16(hi) << 16);
}

// ---- wave reductions --------------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
    return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
    return v;
}

// reduce within 16-lane groups (lanes differing only in bits 0..3)
__device__ __forceinline__ float group16_reduce_sum(float v) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
    return v;
}

__device__ __forceinline__ float group16_reduce_max(float v) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
    return v;
}

// ---- block reduction through LDS (block = N waves, N <= 16) ----------------
template <int MAX_WAVES>
__device__ __forceinline__ float block_reduce_sum(float v, float* lds_scratch) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    v = wave_reduce_sum(v);
    if (lane == 0) lds_scratch[wid] = v;
    __syncthreads();
    const int nwaves = (blockDim.x + WAVE - 1) / WAVE;
    float r = (threadIdx.x < nwaves) ? lds_scratch[threadIdx.x] : 0.0f;
#pragma unroll
    for (int off = MAX_WAVES / 2; off > 0; off >>= 1) r += __shfl_xor(r, off, WAVE);
    // broadcast via lane 0 of wave 0
    if (threadIdx.x == 0) lds_scratch[0] = r;
    __syncthreads();
    return lds_scratch[0];
}

#define HIP_CHECK_LAUNCH()                                                    \
    do {                                                                      \
        hipError_t e_ = hipGetLastError();                                    \
        if (e_ != hipSuccess) return (int)e_;                                 \
    } while (0)

#define CEIL_DIV(a, b) (((a) + (b) - 1) / (b))
