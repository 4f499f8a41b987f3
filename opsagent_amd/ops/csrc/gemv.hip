// Skinny-M GEMM ("GEMV") for decode projections on MI355X:
//   out[M, N] = x[M, K] @ W[N, K]^T      (torch F.linear layout, bf16)
//
// Decode at small batch is weight-streaming-bound: each step reads every
// weight byte once. hipBLASLt's skinny kernels measure ~2.5x off the HBM
// roofline at M=1 (19.9us for a 50 MB qkv read, ~8us at 6.3 TB/s); this
// kernel follows the CDNA guide's GEMV row ("M <= 16 decode weights: operand
// streamed once per block, not shared across waves -> load straight to
// VGPRs, deep unroll, late vmcnt"): W rows stream through 16-B vector loads
// per lane, fp32 accumulate, one wave-reduction per (row, m). The activation
// x is tiny (8-56 KB) and L1-resident after the first pass — each lane
// re-reads only its own 16-B slices, so no LDS staging is needed.
//
// Requires K % 512 == 0 (64 lanes x 8 bf16); callers fall back to hipBLASLt
// otherwise (and for M > 8, where the MFMA path wins).

#include "common.h"
#include <cstdlib>

typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

__device__ __forceinline__ u32x4 nt_load4(const uint32_t* p) {
    return __builtin_nontemporal_load(reinterpret_cast<const u32x4*>(p));
}


// fp32 rstd of x row m: one extra pass over the (L1-resident) activation
__device__ __forceinline__ float row_rstd(const uint32_t* xrow, int k2, int lane,
                                          float eps) {
    float ss = 0.0f;
    for (int i = lane * 4; i < k2; i += WAVE * 4) {
        uint4 xv = *reinterpret_cast<const uint4*>(xrow + i);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            const float lo = bf16_lo((&xv.x)[j]), hi = bf16_hi((&xv.x)[j]);
            ss = fmaf(lo, lo, ss);
            ss = fmaf(hi, hi, ss);
        }
    }
    ss = wave_reduce_sum(ss);
    return rsqrtf(ss / (float)(k2 * 2) + eps);
}

// NORM: x is normalized on the fly (rstd prologue + norm-weight multiply) —
// the separate rmsnorm kernel and its output round-trip disappear.
// ADDRES: the epilogue adds a residual row — out = x @ W^T + residual, i.e.
// the new residual stream is produced directly by the projection.
template <int M, bool NORM, bool ADDRES, int RW = 2>
__global__ __launch_bounds__(256) void gemv_kernel(
    const uint32_t* __restrict__ x,  // [M, K/2]
    const uint32_t* __restrict__ w,  // [N, K/2]
    uint32_t* __restrict__ out,      // [M, N] bf16 (u16 scalar writes)
    const uint32_t* __restrict__ wn, // [K/2] rmsnorm weight (NORM only)
    const uint32_t* __restrict__ res,// [M, N] residual (ADDRES only)
    int N, int k2 /* K/2 */, float eps) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;

    float rstd[M];
    if (NORM) {
#pragma unroll
        for (int m = 0; m < M; ++m) rstd[m] = row_rstd(x + (size_t)m * k2, k2, lane, eps);
    }

    // RW adjacent W rows per wave: multiplies the outstanding 16-B streams
    // per wave (the 1-row form measured 5.1 TB/s vs the 2-row gateup's 6.4)
    for (int row0 = (blockIdx.x * 4 + wid) * RW; row0 < N;
         row0 += gridDim.x * 4 * RW) {
        const uint32_t* wr[RW];
        bool live[RW];
#pragma unroll
        for (int rr_ = 0; rr_ < RW; ++rr_) {
            live[rr_] = row0 + rr_ < N;
            wr[rr_] = w + (size_t)(live[rr_] ? row0 + rr_ : row0) * k2;
        }
        float acc[RW][M];
#pragma unroll
        for (int rr_ = 0; rr_ < RW; ++rr_)
#pragma unroll
            for (int m = 0; m < M; ++m) acc[rr_][m] = 0.0f;
        for (int i = lane * 4; i < k2; i += WAVE * 4) {
            // stream W non-temporally: each byte is read exactly once per
            // step; keep L2 for the KV cache and activations
            u32x4 wv[RW];
#pragma unroll
            for (int rr_ = 0; rr_ < RW; ++rr_) wv[rr_] = nt_load4(wr[rr_] + i);
#pragma unroll
            for (int m = 0; m < M; ++m) {
                uint4 xv = *reinterpret_cast<const uint4*>(x + (size_t)m * k2 + i);
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    float xl = bf16_lo((&xv.x)[j]), xh = bf16_hi((&xv.x)[j]);
                    if (NORM) {
                        const uint32_t wnw = wn[i + j];
                        xl *= rstd[m] * bf16_lo(wnw);
                        xh *= rstd[m] * bf16_hi(wnw);
                    }
#pragma unroll
                    for (int rr_ = 0; rr_ < RW; ++rr_) {
                        acc[rr_][m] = fmaf(xl, bf16_lo(wv[rr_][j]), acc[rr_][m]);
                        acc[rr_][m] = fmaf(xh, bf16_hi(wv[rr_][j]), acc[rr_][m]);
                    }
                }
            }
        }
#pragma unroll
        for (int m = 0; m < M; ++m) {
#pragma unroll
            for (int rr_ = 0; rr_ < RW; ++rr_) {
                float v = wave_reduce_sum(acc[rr_][m]);
                if (lane == 0 && live[rr_]) {
                    if (ADDRES) {
                        v += bf16_to_f32(reinterpret_cast<const uint16_t*>(
                            res)[(size_t)m * N + row0 + rr_]);
                    }
                    reinterpret_cast<uint16_t*>(out)[(size_t)m * N + row0 + rr_] =
                        f32_to_bf16(v);
                }
            }
        }
    }
}

// Fused gate-up + SiLU GEMV for the MLP up-projection: the fused weight
// holds gate rows [0, I) and up rows [I, 2I); each wave computes BOTH rows
// r and I + r and writes act[m][r] = silu(gate) * up directly — the separate
// [1, 2I] intermediate and the silu_mul pass never materialize.
template <int M, bool NORM>
__global__ __launch_bounds__(256) void gemv_gateup_kernel(
    const uint32_t* __restrict__ x,  // [M, K/2]
    const uint32_t* __restrict__ w,  // [2I, K/2] (gate; up)
    uint32_t* __restrict__ out,      // [M, I] bf16
    const uint32_t* __restrict__ wn, // [K/2] rmsnorm weight (NORM only)
    int I, int k2, float eps) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;

    float rstd[M];
    if (NORM) {
#pragma unroll
        for (int m = 0; m < M; ++m) rstd[m] = row_rstd(x + (size_t)m * k2, k2, lane, eps);
    }

    for (int row = blockIdx.x * 4 + wid; row < I; row += gridDim.x * 4) {
        const uint32_t* grow = w + (size_t)row * k2;
        const uint32_t* urow = w + (size_t)(I + row) * k2;
        float accg[M], accu[M];
#pragma unroll
        for (int m = 0; m < M; ++m) accg[m] = accu[m] = 0.0f;
        for (int i = lane * 4; i < k2; i += WAVE * 4) {
            u32x4 gv = nt_load4(grow + i);
            u32x4 uv = nt_load4(urow + i);
#pragma unroll
            for (int m = 0; m < M; ++m) {
                uint4 xv = *reinterpret_cast<const uint4*>(x + (size_t)m * k2 + i);
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    float xl = bf16_lo((&xv.x)[j]), xh = bf16_hi((&xv.x)[j]);
                    if (NORM) {
                        const uint32_t wnw = wn[i + j];
                        xl *= rstd[m] * bf16_lo(wnw);
                        xh *= rstd[m] * bf16_hi(wnw);
                    }
                    accg[m] = fmaf(xl, bf16_lo(gv[j]), accg[m]);
                    accg[m] = fmaf(xh, bf16_hi(gv[j]), accg[m]);
                    accu[m] = fmaf(xl, bf16_lo(uv[j]), accu[m]);
                    accu[m] = fmaf(xh, bf16_hi(uv[j]), accu[m]);
                }
            }
        }
#pragma unroll
        for (int m = 0; m < M; ++m) {
            const float g = wave_reduce_sum(accg[m]);
            const float u = wave_reduce_sum(accu[m]);
            if (lane == 0) {
                const float act = g / (1.0f + __expf(-g)) * u;
                reinterpret_cast<uint16_t*>(out)[(size_t)m * I + row] = f32_to_bf16(act);
            }
        }
    }
}

extern "C" int oa_gemv_gateup_ex(void* stream, const void* x, const void* w,
                                 void* out, const void* wn, int M, int I, int K,
                                 float eps, int norm) {
    if (K % 8 != 0) return -100;
    const int k2 = K / 2;
    const int grid = min(2048, CEIL_DIV(I, 4));
#define LAUNCH_GU(MV)                                                          \
    do {                                                                       \
        if (norm)                                                              \
            hipLaunchKernelGGL((gemv_gateup_kernel<MV, true>), dim3(grid),     \
                               dim3(256), 0, (hipStream_t)stream,              \
                               (const uint32_t*)x, (const uint32_t*)w,         \
                               (uint32_t*)out, (const uint32_t*)wn, I, k2, eps);\
        else                                                                   \
            hipLaunchKernelGGL((gemv_gateup_kernel<MV, false>), dim3(grid),    \
                               dim3(256), 0, (hipStream_t)stream,              \
                               (const uint32_t*)x, (const uint32_t*)w,         \
                               (uint32_t*)out, (const uint32_t*)wn, I, k2, eps);\
    } while (0)
    switch (M) {
        case 1: LAUNCH_GU(1); break;
        case 2: LAUNCH_GU(2); break;
        case 3: LAUNCH_GU(3); break;
        case 4: LAUNCH_GU(4); break;
        case 5: LAUNCH_GU(5); break;
        case 6: LAUNCH_GU(6); break;
        case 7: LAUNCH_GU(7); break;
        case 8: LAUNCH_GU(8); break;
        case 12: LAUNCH_GU(12); break;
        case 16: LAUNCH_GU(16); break;
        default: return -101;
    }
#undef LAUNCH_GU
    HIP_CHECK_LAUNCH();
    return 0;
}

extern "C" int oa_gemv_gateup(void* stream, const void* x, const void* w,
                              void* out, int M, int I, int K) {
    return oa_gemv_gateup_ex(stream, x, w, out, nullptr, M, I, K, 0.0f, 0);
}

// mode: 0 plain, 1 norm-prologue, 2 residual-add epilogue, 3 both
extern "C" int oa_gemv_ex(void* stream, const void* x, const void* w, void* out,
                          const void* wn, const void* res, int M, int N, int K,
                          float eps, int mode) {
    if (K % 8 != 0) return -100;
    const int k2 = K / 2;
    // 4 rows/wave won +3-6% in the ISOLATED probe on wide outputs
    // (scripts/gemv_ab.py) but lost 3% in-model (decode step 3.63 -> 3.75 ms,
    // within-box A/B) — real decode interleaves the W stream with KV/L2
    // traffic the probe does not. Default OFF; OPSAGENT_GEMV_RW4_MIN_N
    // re-enables for experiments.
    static int rw4_min_n = -1;
    static int rw1_max_n = -1;
    if (rw4_min_n < 0) {
        const char* e = getenv("OPSAGENT_GEMV_RW4_MIN_N");
        rw4_min_n = e ? atoi(e) : (1 << 30);
        // RW=1 for small-N outputs: at N=4096 the RW=2 grid is only 512
        // workgroups (2 per CU — half the SIMD slots idle); one row per wave
        // doubles the wave count and the streams in flight. In-model A/B
        // (8B turn, same box): off 438.1 ms, N<=6144 431.0 ms, N<=4096
        // 431.9 ms — default N<=6144 (covers o_proj and the fused-norm qkv).
        const char* e1 = getenv("OPSAGENT_GEMV_RW1_MAX_N");
        rw1_max_n = e1 ? atoi(e1) : 6144;
    }
    const bool rw4 = N >= rw4_min_n;
    const bool rw1 = !rw4 && N <= rw1_max_n;
    const int grid = min(2048, CEIL_DIV(N, rw4 ? 16 : (rw1 ? 4 : 8)));
#define LAUNCH_NM(MV, NORMV, RESV)                                            \
    do {                                                                       \
        if (rw4)                                                               \
            hipLaunchKernelGGL((gemv_kernel<MV, NORMV, RESV, 4>), dim3(grid),  \
                               dim3(256), 0, (hipStream_t)stream,              \
                               (const uint32_t*)x, (const uint32_t*)w,         \
                               (uint32_t*)out, (const uint32_t*)wn,            \
                               (const uint32_t*)res, N, k2, eps);              \
        else if (rw1)                                                          \
            hipLaunchKernelGGL((gemv_kernel<MV, NORMV, RESV, 1>), dim3(grid),  \
                               dim3(256), 0, (hipStream_t)stream,              \
                               (const uint32_t*)x, (const uint32_t*)w,         \
                               (uint32_t*)out, (const uint32_t*)wn,            \
                               (const uint32_t*)res, N, k2, eps);              \
        else                                                                   \
            hipLaunchKernelGGL((gemv_kernel<MV, NORMV, RESV, 2>), dim3(grid),  \
                               dim3(256), 0, (hipStream_t)stream,              \
                               (const uint32_t*)x, (const uint32_t*)w,         \
                               (uint32_t*)out, (const uint32_t*)wn,            \
                               (const uint32_t*)res, N, k2, eps);              \
    } while (0)
#define LAUNCH_MODE(MV)                                                        \
    do {                                                                       \
        switch (mode) {                                                        \
            case 0: LAUNCH_NM(MV, false, false); break;                        \
            case 1: LAUNCH_NM(MV, true, false); break;                         \
            case 2: LAUNCH_NM(MV, false, true); break;                         \
            case 3: LAUNCH_NM(MV, true, true); break;                          \
            default: return -102;                                              \
        }                                                                      \
    } while (0)
    switch (M) {
        case 1: LAUNCH_MODE(1); break;
        case 2: LAUNCH_MODE(2); break;
        case 3: LAUNCH_MODE(3); break;
        case 4: LAUNCH_MODE(4); break;
        case 5: LAUNCH_MODE(5); break;
        case 6: LAUNCH_MODE(6); break;
        case 7: LAUNCH_MODE(7); break;
        case 8: LAUNCH_MODE(8); break;
        case 12: LAUNCH_MODE(12); break;
        case 16: LAUNCH_MODE(16); break;
        default: return -101;
    }
#undef LAUNCH_MODE
#undef LAUNCH_NM
    HIP_CHECK_LAUNCH();
    return 0;
}

extern "C" int oa_gemv(void* stream, const void* x, const void* w, void* out,
                       int M, int N, int K) {
    return oa_gemv_ex(stream, x, w, out, nullptr, nullptr, M, N, K, 0.0f, 0);
}

// RW=4 probe variant (within-probe A/B only — guide rule 24)
extern "C" int oa_gemv_rw4(void* stream, const void* x, const void* w, void* out,
                           int M, int N, int K) {
    if (K % 8 != 0) return -100;
    const int k2 = K / 2;
    const int grid = min(2048, CEIL_DIV(N, 16));
    switch (M) {
        case 1:
            hipLaunchKernelGGL((gemv_kernel<1, false, false, 4>), dim3(grid),
                               dim3(256), 0, (hipStream_t)stream,
                               (const uint32_t*)x, (const uint32_t*)w,
                               (uint32_t*)out, nullptr, nullptr, N, k2, 0.0f);
            break;
        default:
            return -101;
    }
    HIP_CHECK_LAUNCH();
    return 0;
}
