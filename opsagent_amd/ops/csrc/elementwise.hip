// Element-wise fused kernels: SwiGLU activation (silu(gate) * up) and the
// paged-KV scatter write. Memory-bound; bf16x8 vectorized per guide G13.

#include "common.h"

// gate/up/out: flat [N] bf16, N % 8 == 0
__global__ __launch_bounds__(256) void silu_mul_kernel(
    const uint32_t* __restrict__ gate, const uint32_t* __restrict__ up,
    uint32_t* __restrict__ out, int64_t n2 /* N/2 words */) {
    for (int64_t i = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) * 4; i < n2;
         i += (int64_t)gridDim.x * blockDim.x * 4) {
        uint4 g = *reinterpret_cast<const uint4*>(gate + i);
        uint4 u = *reinterpret_cast<const uint4*>(up + i);
        uint4 o;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            uint32_t gw = (&g.x)[j], uw = (&u.x)[j];
            float glo = bf16_lo(gw), ghi = bf16_hi(gw);
            float lo = glo / (1.0f + __expf(-glo)) * bf16_lo(uw);
            float hi = ghi / (1.0f + __expf(-ghi)) * bf16_hi(uw);
            (&o.x)[j] = pack_bf16x2(lo, hi);
        }
        *reinterpret_cast<uint4*>(out + i) = o;
    }
}

extern "C" int oa_silu_mul(void* stream, const void* gate, const void* up,
                           void* out, int64_t n) {
    if (n % 8 != 0) return -100;
    int64_t n2 = n / 2;
    int grid = (int)min((int64_t)2048, CEIL_DIV(n2 / 4, 256));
    hipLaunchKernelGGL(silu_mul_kernel, dim3(grid), dim3(256), 0,
                       (hipStream_t)stream, (const uint32_t*)gate,
                       (const uint32_t*)up, (uint32_t*)out, n2);
    HIP_CHECK_LAUNCH();
    return 0;
}

// Scatter new K/V token rows into the paged cache.
// k/v: [T, Hk, D] bf16; caches viewed flat as [num_slots, Hk, D];
// slot_mapping: [T] int32 flat slot index (block_id * block_size + offset).
__global__ __launch_bounds__(256) void kv_write_kernel(
    uint32_t* __restrict__ kc, uint32_t* __restrict__ vc,
    const uint32_t* __restrict__ k, const uint32_t* __restrict__ v,
    const int* __restrict__ slots, int T, int row_words /* Hk*D/2 */) {
    const int64_t total = (int64_t)T * (row_words / 4);
    for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
         idx += (int64_t)gridDim.x * blockDim.x) {
        const int t = idx / (row_words / 4);
        const int w = (idx % (row_words / 4)) * 4;
        const int64_t src = (int64_t)t * row_words + w;
        const int64_t dst = (int64_t)slots[t] * row_words + w;
        *reinterpret_cast<uint4*>(kc + dst) = *reinterpret_cast<const uint4*>(k + src);
        *reinterpret_cast<uint4*>(vc + dst) = *reinterpret_cast<const uint4*>(v + src);
    }
}

extern "C" int oa_kv_write(void* stream, void* k_cache, void* v_cache,
                           const void* k, const void* v, const void* slots,
                           int T, int hk, int d) {
    const int row_words = hk * d / 2;
    if ((hk * d) % 8 != 0) return -100;
    int64_t total = (int64_t)T * (row_words / 4);
    int grid = (int)min((int64_t)2048, CEIL_DIV(total, 256));
    hipLaunchKernelGGL(kv_write_kernel, dim3(grid), dim3(256), 0,
                       (hipStream_t)stream, (uint32_t*)k_cache, (uint32_t*)v_cache,
                       (const uint32_t*)k, (const uint32_t*)v, (const int*)slots,
                       T, row_words);
    HIP_CHECK_LAUNCH();
    return 0;
}
