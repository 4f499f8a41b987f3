// Grouped MoE expert GEMV for decode/small-batch on MI355X.
//
// The per-expert python loop costs ~2 kernel launches x active experts x
// MoE layers per token; these kernels do the whole layer stage in ONE
// launch with expert indirection: pair p = (token_ids[p], expert_ids[p])
// computes against expert expert_ids[p]'s weights.
//
//   oa_moe_gateup[_fp8]: act[p, :I] = silu(x[tok] @ gate_e^T) * (x[tok] @ up_e^T)
//   oa_moe_down[_fp8]:   y[p, :H]  = (act[p] @ w2_e^T) * pair_weight[p]
//
// The caller then out.index_add_(0, token_ids, y) — one scatter per layer.
// Weights: bf16 [E, 2I|H, K] or fp8 [E, 2I|H, K] + per-row f32 scales.

#include "common.h"

typedef __attribute__((ext_vector_type(4))) unsigned int u32x4m;
typedef __attribute__((ext_vector_type(2))) float f32x2m;

__device__ __forceinline__ u32x4m nt_ld4(const uint32_t* p) {
    return __builtin_nontemporal_load(reinterpret_cast<const u32x4m*>(p));
}

template <bool WORD>
__device__ __forceinline__ f32x2m f8x2_f32(unsigned int src) {
    return __builtin_amdgcn_cvt_pk_f32_fp8(src, WORD);
}

// ---- shared row-dot helpers -------------------------------------------------
// dot of bf16 row (w) against bf16 x row; k2 = K/2 words
__device__ __forceinline__ float dot_bf16(const uint32_t* wrow, const uint32_t* xrow,
                                          int k2, int lane) {
    float acc = 0.0f;
    for (int i = lane * 4; i < k2; i += WAVE * 4) {
        u32x4m wv = nt_ld4(wrow + i);
        uint4 xv = *reinterpret_cast<const uint4*>(xrow + i);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            acc = fmaf(bf16_lo((&xv.x)[j]), bf16_lo(wv[j]), acc);
            acc = fmaf(bf16_hi((&xv.x)[j]), bf16_hi(wv[j]), acc);
        }
    }
    return wave_reduce_sum(acc);
}

// dot of fp8 row against bf16 x row; k4 = K/4 words
__device__ __forceinline__ float dot_fp8(const uint32_t* wrow, const uint32_t* xrow,
                                         int k4, int lane) {
    float acc = 0.0f;
    for (int i = lane * 4; i < k4; i += WAVE * 4) {
        u32x4m wv = nt_ld4(wrow + i);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            f32x2m lo = f8x2_f32<false>(wv[j]);
            f32x2m hi = f8x2_f32<true>(wv[j]);
            uint2 xv = *reinterpret_cast<const uint2*>(xrow + (i + j) * 2);
            acc = fmaf(bf16_lo(xv.x), lo[0], acc);
            acc = fmaf(bf16_hi(xv.x), lo[1], acc);
            acc = fmaf(bf16_lo(xv.y), hi[0], acc);
            acc = fmaf(bf16_hi(xv.y), hi[1], acc);
        }
    }
    return wave_reduce_sum(acc);
}

// ---- gate-up + SiLU ---------------------------------------------------------
template <bool FP8>
__global__ __launch_bounds__(256) void moe_gateup_kernel(
    const uint32_t* __restrict__ x,       // [T, K/2] bf16
    const uint32_t* __restrict__ w,       // [E, 2I, K] bf16 or fp8 words
    const float* __restrict__ wscale,     // [E, 2I] (fp8) or nullptr
    const int* __restrict__ expert_ids,   // [P]
    const int* __restrict__ token_ids,    // [P]
    uint32_t* __restrict__ act,           // [P, I] bf16
    int I, int K) {
    const int p = blockIdx.y;
    const int e = expert_ids[p];
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int wwords = FP8 ? K / 4 : K / 2;
    const uint32_t* xrow = x + (size_t)token_ids[p] * (K / 2);
    const uint32_t* wbase = w + (size_t)e * 2 * I * wwords;
    const float* sbase = FP8 ? wscale + (size_t)e * 2 * I : nullptr;

    for (int row = blockIdx.x * 4 + wid; row < I; row += gridDim.x * 4) {
        float g, u;
        if (FP8) {
            g = dot_fp8(wbase + (size_t)row * wwords, xrow, wwords, lane) * sbase[row];
            u = dot_fp8(wbase + (size_t)(I + row) * wwords, xrow, wwords, lane) *
                sbase[I + row];
        } else {
            g = dot_bf16(wbase + (size_t)row * wwords, xrow, wwords, lane);
            u = dot_bf16(wbase + (size_t)(I + row) * wwords, xrow, wwords, lane);
        }
        if (lane == 0) {
            const float a = g / (1.0f + __expf(-g)) * u;
            reinterpret_cast<uint16_t*>(act)[(size_t)p * I + row] = f32_to_bf16(a);
        }
    }
}

// ---- down projection (scaled by routing weight) ----------------------------
template <bool FP8>
__global__ __launch_bounds__(256) void moe_down_kernel(
    const uint32_t* __restrict__ act,     // [P, I/2] bf16
    const uint32_t* __restrict__ w,       // [E, H, I] bf16 or fp8
    const float* __restrict__ wscale,     // [E, H] or nullptr
    const int* __restrict__ expert_ids,   // [P]
    const float* __restrict__ pair_w,     // [P] routing weights
    uint32_t* __restrict__ y,             // [P, H] bf16
    int H, int I) {
    const int p = blockIdx.y;
    const int e = expert_ids[p];
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int wwords = FP8 ? I / 4 : I / 2;
    const uint32_t* arow = act + (size_t)p * (I / 2);
    const uint32_t* wbase = w + (size_t)e * H * wwords;
    const float* sbase = FP8 ? wscale + (size_t)e * H : nullptr;
    const float pw = pair_w[p];

    for (int row = blockIdx.x * 4 + wid; row < H; row += gridDim.x * 4) {
        float v;
        if (FP8)
            v = dot_fp8(wbase + (size_t)row * wwords, arow, wwords, lane) * sbase[row];
        else
            v = dot_bf16(wbase + (size_t)row * wwords, arow, wwords, lane);
        if (lane == 0)
            reinterpret_cast<uint16_t*>(y)[(size_t)p * H + row] = f32_to_bf16(v * pw);
    }
}

// ---- expert-major variants (big-batch decode) ---------------
// One block column per EXPERT: the block scans the pair list once into LDS,
// then streams the expert's weight rows ONCE PER GROUP of up to 8 matched
// pairs (the pair-major kernels above re-stream the full expert weights for
// every pair — linear blowup with batch). Shapes are static in (E, P), so
// the whole stage stays hipGraph-capturable at any decode batch.

#define EMAJ_MAXP 2048   // pair-list capacity (batch 256 x top-k 8)
#define EMAJ_GROUP 8     // pairs per weight stream

template <bool FP8, bool GATEUP>
__global__ __launch_bounds__(256) void moe_emaj_kernel(
    const uint32_t* __restrict__ x,       // [T, K/2] bf16 (or act [P, I/2])
    const uint32_t* __restrict__ w,       // [E, 2I|H, K] bf16/fp8 words
    const float* __restrict__ wscale,     // [E, 2I|H] or nullptr
    const int* __restrict__ expert_ids,   // [P]
    const int* __restrict__ token_ids,    // [P] (GATEUP; ignored for down)
    const float* __restrict__ pair_w,     // [P] (down; ignored for gateup)
    uint32_t* __restrict__ out,           // [P, I] or [P, H] bf16
    int P, int NROWS /* I or H */, int K) {
    const int e = blockIdx.y;
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;

    __shared__ int plist[EMAJ_MAXP];
    __shared__ int pcount;
    if (threadIdx.x == 0) pcount = 0;
    __syncthreads();
    for (int p = threadIdx.x; p < P; p += blockDim.x)
        if (expert_ids[p] == e) {
            const int slot = atomicAdd(&pcount, 1);
            if (slot < EMAJ_MAXP) plist[slot] = p;
        }
    __syncthreads();
    const int cnt = min(pcount, EMAJ_MAXP);
    if (cnt == 0) return;

    const int wwords = FP8 ? K / 4 : K / 2;
    const int xwords = K / 2;
    const uint32_t* wbase =
        w + (size_t)e * (GATEUP ? 2 * NROWS : NROWS) * wwords;
    const float* sbase =
        FP8 ? wscale + (size_t)e * (GATEUP ? 2 * NROWS : NROWS) : nullptr;

    for (int row = blockIdx.x * 4 + wid; row < NROWS; row += gridDim.x * 4) {
        const uint32_t* wrow_g = wbase + (size_t)row * wwords;
        const uint32_t* wrow_u =
            GATEUP ? wbase + (size_t)(NROWS + row) * wwords : nullptr;
        for (int g0 = 0; g0 < cnt; g0 += EMAJ_GROUP) {
            const int m = min(EMAJ_GROUP, cnt - g0);
            // all register arrays are indexed by the UNROLLED c only (a
            // runtime index would demote them to scratch — guide rule 20)
            float accg[EMAJ_GROUP], accu[EMAJ_GROUP];
            const uint32_t* xbase[EMAJ_GROUP];
#pragma unroll
            for (int c = 0; c < EMAJ_GROUP; ++c) {
                accg[c] = accu[c] = 0.0f;
                int p = plist[g0];  // safe default
                if (c < m) p = plist[g0 + c];
                xbase[c] = x + (size_t)(GATEUP ? token_ids[p] : p) * xwords;
            }
            // stream the weight row(s) once; dot against every pair's x row
            for (int i = lane * 4; i < wwords; i += WAVE * 4) {
                u32x4m wg = nt_ld4(wrow_g + i);
                u32x4m wu{};
                if (GATEUP) wu = nt_ld4(wrow_u + i);
                float wfg[16], wfu[16];
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    if (FP8) {
                        f32x2m lo = f8x2_f32<false>(wg[j]);
                        f32x2m hi = f8x2_f32<true>(wg[j]);
                        wfg[j * 4] = lo[0]; wfg[j * 4 + 1] = lo[1];
                        wfg[j * 4 + 2] = hi[0]; wfg[j * 4 + 3] = hi[1];
                        if (GATEUP) {
                            f32x2m lo2 = f8x2_f32<false>(wu[j]);
                            f32x2m hi2 = f8x2_f32<true>(wu[j]);
                            wfu[j * 4] = lo2[0]; wfu[j * 4 + 1] = lo2[1];
                            wfu[j * 4 + 2] = hi2[0]; wfu[j * 4 + 3] = hi2[1];
                        }
                    } else {
                        wfg[j * 2] = bf16_lo(wg[j]);
                        wfg[j * 2 + 1] = bf16_hi(wg[j]);
                        if (GATEUP) {
                            wfu[j * 2] = bf16_lo(wu[j]);
                            wfu[j * 2 + 1] = bf16_hi(wu[j]);
                        }
                    }
                }
                const int xoff = FP8 ? i * 2 : i;
#pragma unroll
                for (int c = 0; c < EMAJ_GROUP; ++c) {
                    if (c >= m) continue;
                    const uint32_t* xr = xbase[c] + xoff;
#pragma unroll
                    for (int j = 0; j < (FP8 ? 8 : 4); ++j) {
                        const uint32_t xv = xr[j];
                        accg[c] = fmaf(bf16_lo(xv), wfg[j * 2], accg[c]);
                        accg[c] = fmaf(bf16_hi(xv), wfg[j * 2 + 1], accg[c]);
                        if (GATEUP) {
                            accu[c] = fmaf(bf16_lo(xv), wfu[j * 2], accu[c]);
                            accu[c] = fmaf(bf16_hi(xv), wfu[j * 2 + 1], accu[c]);
                        }
                    }
                }
            }
#pragma unroll
            for (int c = 0; c < EMAJ_GROUP; ++c) {
                if (c >= m) continue;
                const int p = plist[g0 + c];
                float g = wave_reduce_sum(accg[c]);
                float u = GATEUP ? wave_reduce_sum(accu[c]) : 0.0f;
                if (lane == 0) {
                    if (GATEUP) {
                        if (FP8) {
                            g *= sbase[row];
                            u *= sbase[NROWS + row];
                        }
                        const float a = g / (1.0f + __expf(-g)) * u;
                        reinterpret_cast<uint16_t*>(out)[(size_t)p * NROWS + row] =
                            f32_to_bf16(a);
                    } else {
                        if (FP8) g *= sbase[row];
                        reinterpret_cast<uint16_t*>(out)[(size_t)p * NROWS + row] =
                            f32_to_bf16(g * pair_w[p]);
                    }
                }
            }
        }
    }
}

extern "C" int oa_moe_gateup_emaj(void* stream, const void* x, const void* w,
                                  const void* wscale, const void* expert_ids,
                                  const void* token_ids, void* act, int P,
                                  int E, int I, int K, int fp8) {
    if (K % 16 != 0 || P > EMAJ_MAXP) return -100;
    dim3 grid(min(64, CEIL_DIV(I, 4)), E), block(256);
    if (fp8)
        hipLaunchKernelGGL((moe_emaj_kernel<true, true>), grid, block, 0,
                           (hipStream_t)stream, (const uint32_t*)x,
                           (const uint32_t*)w, (const float*)wscale,
                           (const int*)expert_ids, (const int*)token_ids,
                           (const float*)nullptr, (uint32_t*)act, P, I, K);
    else
        hipLaunchKernelGGL((moe_emaj_kernel<false, true>), grid, block, 0,
                           (hipStream_t)stream, (const uint32_t*)x,
                           (const uint32_t*)w, (const float*)wscale,
                           (const int*)expert_ids, (const int*)token_ids,
                           (const float*)nullptr, (uint32_t*)act, P, I, K);
    HIP_CHECK_LAUNCH();
    return 0;
}

extern "C" int oa_moe_down_emaj(void* stream, const void* act, const void* w,
                                const void* wscale, const void* expert_ids,
                                const void* pair_w, void* y, int P, int E,
                                int H, int I, int fp8) {
    if (I % 16 != 0 || P > EMAJ_MAXP) return -100;
    dim3 grid(min(64, CEIL_DIV(H, 4)), E), block(256);
    if (fp8)
        hipLaunchKernelGGL((moe_emaj_kernel<true, false>), grid, block, 0,
                           (hipStream_t)stream, (const uint32_t*)act,
                           (const uint32_t*)w, (const float*)wscale,
                           (const int*)expert_ids, (const int*)nullptr,
                           (const float*)pair_w, (uint32_t*)y, P, H, I);
    else
        hipLaunchKernelGGL((moe_emaj_kernel<false, false>), grid, block, 0,
                           (hipStream_t)stream, (const uint32_t*)act,
                           (const uint32_t*)w, (const float*)wscale,
                           (const int*)expert_ids, (const int*)nullptr,
                           (const float*)pair_w, (uint32_t*)y, P, H, I);
    HIP_CHECK_LAUNCH();
    return 0;
}

extern "C" int oa_moe_gateup(void* stream, const void* x, const void* w,
                             const void* wscale, const void* expert_ids,
                             const void* token_ids, void* act, int P, int I,
                             int K, int fp8) {
    if (K % 16 != 0) return -100;
    dim3 grid(min(512, CEIL_DIV(I, 4)), P), block(256);
    if (fp8)
        hipLaunchKernelGGL((moe_gateup_kernel<true>), grid, block, 0,
                           (hipStream_t)stream, (const uint32_t*)x,
                           (const uint32_t*)w, (const float*)wscale,
                           (const int*)expert_ids, (const int*)token_ids,
                           (uint32_t*)act, I, K);
    else
        hipLaunchKernelGGL((moe_gateup_kernel<false>), grid, block, 0,
                           (hipStream_t)stream, (const uint32_t*)x,
                           (const uint32_t*)w, (const float*)wscale,
                           (const int*)expert_ids, (const int*)token_ids,
                           (uint32_t*)act, I, K);
    HIP_CHECK_LAUNCH();
    return 0;
}

extern "C" int oa_moe_down(void* stream, const void* act, const void* w,
                           const void* wscale, const void* expert_ids,
                           const void* pair_w, void* y, int P, int H, int I,
                           int fp8) {
    if (I % 16 != 0) return -100;
    dim3 grid(min(512, CEIL_DIV(H, 4)), P), block(256);
    if (fp8)
        hipLaunchKernelGGL((moe_down_kernel<true>), grid, block, 0,
                           (hipStream_t)stream, (const uint32_t*)act,
                           (const uint32_t*)w, (const float*)wscale,
                           (const int*)expert_ids, (const float*)pair_w,
                           (uint32_t*)y, H, I);
    else
        hipLaunchKernelGGL((moe_down_kernel<false>), grid, block, 0,
                           (hipStream_t)stream, (const uint32_t*)act,
                           (const uint32_t*)w, (const float*)wscale,
                           (const int*)expert_ids, (const float*)pair_w,
                           (uint32_t*)y, H, I);
    HIP_CHECK_LAUNCH();
    return 0;
}
