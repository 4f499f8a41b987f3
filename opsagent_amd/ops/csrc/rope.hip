// RoPE (neox/rotate-half) kernel for MI355X — applies rotary embedding
// in-place to Q and K after the QKV projection.
//
// Per CDNA guide Appendix B (element-wise / trig-heavy): the cos/sin tables
// are precomputed on the HOST ([max_seq, D/2] f32) — on-device sinf/cosf
// turns a memory-bound op VALU-bound (guide: 16% of HBM BW). Loads are
// vectorized (4 bf16 = 8 B per half, plus 16 B of f32 table per lane).
//
// Replaces SURVEY.md §2b "RoPE kernel".

#include "common.h"

// x: [T, H, D] bf16, table indexed by positions[t]. Work item = (t, h, 4 dims).
__global__ __launch_bounds__(256) void rope_kernel(
    uint32_t* __restrict__ x,           // [T, H, D/2] packed bf16x2 view
    const float* __restrict__ cs,       // [max_seq, D/2] cos
    const float* __restrict__ sn,       // [max_seq, D/2] sin
    const int* __restrict__ positions,  // [T]
    int T, int H, int d2 /* D/2 */) {
    // each work item covers 4 dim-pairs: loads 2 words (4 bf16) from each half
    const int per_row = d2 / 4;  // work items per (t,h)
    const int64_t total = (int64_t)T * H * per_row;
    for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
         idx += (int64_t)gridDim.x * blockDim.x) {
        const int it = idx / (H * per_row);
        const int rem = idx % (H * per_row);
        const int h = rem / per_row;
        const int dblk = (rem % per_row) * 4;  // first dim-pair index
        const int pos = positions[it];

        uint32_t* row = x + ((size_t)it * H + h) * d2;  // d2 words = D bf16
        // first half at word offset dblk/2, second half at (d2 + dblk)/2
        uint2 w1 = *reinterpret_cast<uint2*>(row + dblk / 2);
        uint2 w2 = *reinterpret_cast<uint2*>(row + (d2 + dblk) / 2);
        float4 c = *reinterpret_cast<const float4*>(cs + (size_t)pos * d2 + dblk);
        float4 s = *reinterpret_cast<const float4*>(sn + (size_t)pos * d2 + dblk);

        float x1[4] = {bf16_lo(w1.x), bf16_hi(w1.x), bf16_lo(w1.y), bf16_hi(w1.y)};
        float x2[4] = {bf16_lo(w2.x), bf16_hi(w2.x), bf16_lo(w2.y), bf16_hi(w2.y)};
        float o1[4], o2[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            float cj = (&c.x)[j], sj = (&s.x)[j];
            o1[j] = x1[j] * cj - x2[j] * sj;
            o2[j] = x2[j] * cj + x1[j] * sj;
        }
        uint2 ow1, ow2;
        ow1.x = pack_bf16x2(o1[0], o1[1]);
        ow1.y = pack_bf16x2(o1[2], o1[3]);
        ow2.x = pack_bf16x2(o2[0], o2[1]);
        ow2.y = pack_bf16x2(o2[2], o2[3]);
        *reinterpret_cast<uint2*>(row + dblk / 2) = ow1;
        *reinterpret_cast<uint2*>(row + (d2 + dblk) / 2) = ow2;
    }
}

extern "C" int oa_rope(void* stream, void* q, void* k, const void* cos_t,
                       const void* sin_t, const void* positions, int T, int Hq,
                       int Hk, int D) {
    if (D % 16 != 0) return -100;
    const int d2 = D / 2;
    // grid sized per guide G11: cap ~2048 blocks, grid-stride the rest
    int64_t items_q = (int64_t)T * Hq * (d2 / 4);
    int grid_q = (int)min((int64_t)2048, CEIL_DIV(items_q, 256));
    hipLaunchKernelGGL(rope_kernel, dim3(grid_q), dim3(256), 0, (hipStream_t)stream,
                       (uint32_t*)q, (const float*)cos_t, (const float*)sin_t,
                       (const int*)positions, T, Hq, d2);
    HIP_CHECK_LAUNCH();
    int64_t items_k = (int64_t)T * Hk * (d2 / 4);
    int grid_k = (int)min((int64_t)2048, CEIL_DIV(items_k, 256));
    hipLaunchKernelGGL(rope_kernel, dim3(grid_k), dim3(256), 0, (hipStream_t)stream,
                       (uint32_t*)k, (const float*)cos_t, (const float*)sin_t,
                       (const int*)positions, T, Hk, d2);
    HIP_CHECK_LAUNCH();
    return 0;
}
