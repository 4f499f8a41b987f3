// RMSNorm kernels for MI355X (gfx950) — memory-bound, HBM-roofline targets.
//
// Replaces the vacant model layer of the reference (SURVEY.md §2b "RMSNorm
// kernel (fused residual-add variant)"). Design per CDNA guide Appendix B:
// one 256-thread block per row batch, bf16x8 (16 B/lane) vectorized loads
// (guide: scalar bf16 2.35 TB/s -> vectorized 4.89 TB/s), f32 accumulation,
// wave + LDS block reduce, fused residual-add variant writes both the normed
// output and the new residual stream in one pass.

#include "common.h"

// one block per row; D multiple of 8; BLOCK=256 threads
template <bool FUSED_ADD>
__global__ __launch_bounds__(256) void rmsnorm_kernel(
    const uint32_t* __restrict__ x,   // [rows, D/2] packed bf16x2
    uint32_t* __restrict__ residual,  // [rows, D/2] in/out (FUSED_ADD only)
    const uint32_t* __restrict__ w,   // [D/2]
    uint32_t* __restrict__ out,       // [rows, D/2]
    int rows, int d2 /* = D/2 */, float inv_d, float eps) {
    const int row = blockIdx.x;
    if (row >= rows) return;
    const uint32_t* xrow = x + (size_t)row * d2;
    uint32_t* rrow = FUSED_ADD ? residual + (size_t)row * d2 : nullptr;
    uint32_t* orow = out + (size_t)row * d2;

    __shared__ float red[16];
    // 32 f32 = 16 bf16x2 words per thread: covers D <= 8192 at 256 threads
    float vals[32];
    int nw = 0;
    float sumsq = 0.0f;
    for (int i = threadIdx.x * 4; i < d2; i += blockDim.x * 4) {
        // 16-byte vector load: 4 words = 8 bf16
        uint4 v = *reinterpret_cast<const uint4*>(xrow + i);
        uint4 r;
        if (FUSED_ADD) r = *reinterpret_cast<const uint4*>(rrow + i);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            uint32_t word = (&v.x)[j];
            float lo = bf16_lo(word), hi = bf16_hi(word);
            if (FUSED_ADD) {
                uint32_t rw = (&r.x)[j];
                lo += bf16_lo(rw);
                hi += bf16_hi(rw);
            }
            vals[nw * 2 + 0] = lo;
            vals[nw * 2 + 1] = hi;
            ++nw;
            sumsq += lo * lo + hi * hi;
        }
    }
    if (FUSED_ADD) {
        // write the new residual stream (x + residual)
        int k = 0;
        for (int i = threadIdx.x * 4; i < d2; i += blockDim.x * 4) {
            uint4 o;
#pragma unroll
            for (int j = 0; j < 4; ++j)
                (&o.x)[j] = pack_bf16x2(vals[(k + j) * 2], vals[(k + j) * 2 + 1]);
            *reinterpret_cast<uint4*>(rrow + i) = o;
            k += 4;
        }
    }
    float total = block_reduce_sum<16>(sumsq, red);
    const float scale = rsqrtf(total * inv_d + eps);

    int k = 0;
    for (int i = threadIdx.x * 4; i < d2; i += blockDim.x * 4) {
        uint4 wv = *reinterpret_cast<const uint4*>(w + i);
        uint4 o;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            uint32_t ww = (&wv.x)[j];
            float lo = vals[(k + j) * 2] * scale * bf16_lo(ww);
            float hi = vals[(k + j) * 2 + 1] * scale * bf16_hi(ww);
            (&o.x)[j] = pack_bf16x2(lo, hi);
        }
        *reinterpret_cast<uint4*>(orow + i) = o;
        k += 4;
    }
}

extern "C" int oa_rmsnorm(void* stream, const void* x, const void* w, void* out,
                          int rows, int dim, float eps) {
    if (dim % 8 != 0 || dim > 8192) return -100;
    dim3 grid(rows), block(256);
    hipLaunchKernelGGL((rmsnorm_kernel<false>), grid, block, 0, (hipStream_t)stream,
                       (const uint32_t*)x, nullptr, (const uint32_t*)w, (uint32_t*)out,
                       rows, dim / 2, 1.0f / dim, eps);
    HIP_CHECK_LAUNCH();
    return 0;
}

extern "C" int oa_fused_add_rmsnorm(void* stream, const void* x, void* residual,
                                    const void* w, void* out, int rows, int dim,
                                    float eps) {
    if (dim % 8 != 0 || dim > 8192) return -100;
    dim3 grid(rows), block(256);
    hipLaunchKernelGGL((rmsnorm_kernel<true>), grid, block, 0, (hipStream_t)stream,
                       (const uint32_t*)x, (uint32_t*)residual, (const uint32_t*)w,
                       (uint32_t*)out, rows, dim / 2, 1.0f / dim, eps);
    HIP_CHECK_LAUNCH();
    return 0;
}
