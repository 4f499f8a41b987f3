// FP8 (OCP e4m3) expert-GEMM path for MoE on MI355X — SURVEY.md §2b "GEMM
// path: ... custom fp8 MFMA for MoE experts".
//
// gfx950 fp8 is OCP e4m3fn (NOT MI300X fnuz — CDNA guide §4). Weights are
// quantized per output row (scale = absmax/448); activations per token.
//
// Three kernels:
//   oa_quant_fp8    bf16 [T, K] -> fp8 [T, K] + f32 scales [T]
//   oa_gemv_fp8     skinny M <= 8 decode path: fp8 W stream (HALF the bytes
//                   of bf16 -> ~2x decode speed for expert projections),
//                   bf16 x, in-register HW dequant (v_cvt_pk_f32_fp8)
//   oa_gemm_fp8     MFMA tile GEMM for prefill-sized M:
//                   C[M,N] = (A8 @ B8^T) * a_scale[m] * b_scale[n] on
//                   v_mfma_f32_16x16x32_fp8_fp8, 128x128 LDS-staged tiles
//                   (the CDNA guide ladder's step-2 structure)

#include "common.h"

typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;
typedef __attribute__((ext_vector_type(2))) int i32x2;

__device__ __forceinline__ u32x4 nt_load4f(const uint32_t* p) {
    return __builtin_nontemporal_load(reinterpret_cast<const u32x4*>(p));
}

#define FP8_MAX 448.0f

// convert 2 packed fp8 (low/high half-word of src selected by WORD) to floats
template <bool WORD>
__device__ __forceinline__ f32x2 fp8x2_to_f32(unsigned int src) {
    return __builtin_amdgcn_cvt_pk_f32_fp8(src, WORD);
}

// pack two floats into 2 fp8 within `old`'s selected half-word
template <bool WORD>
__device__ __forceinline__ unsigned int f32x2_to_fp8(float a, float b,
                                                     unsigned int old) {
    return __builtin_amdgcn_cvt_pk_fp8_f32(a, b, old, WORD);
}

// ---- per-token quantization -------------------------------------------------
// one wave per row; 64 lanes x 8 elems per pass
__global__ __launch_bounds__(256) void quant_fp8_kernel(
    const uint32_t* __restrict__ x,  // [T, K/2] bf16x2
    uint32_t* __restrict__ q,        // [T, K/4] fp8x4
    float* __restrict__ scales,      // [T]
    int T, int K) {
    const int row = blockIdx.x * 4 + threadIdx.x / WAVE;
    if (row >= T) return;
    const int lane = threadIdx.x & (WAVE - 1);
    const int k2 = K / 2;
    const uint32_t* xr = x + (size_t)row * k2;
    // pass 1: absmax
    float amax = 1e-8f;
    for (int i = lane * 4; i < k2; i += WAVE * 4) {
        uint4 w = *reinterpret_cast<const uint4*>(xr + i);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            amax = fmaxf(amax, fabsf(bf16_lo((&w.x)[j])));
            amax = fmaxf(amax, fabsf(bf16_hi((&w.x)[j])));
        }
    }
    amax = wave_reduce_max(amax);
    const float scale = amax / FP8_MAX;
    const float inv = FP8_MAX / amax;
    if (lane == 0) scales[row] = scale;
    // pass 2: quantize 8 bf16 -> 2 fp8 words per lane chunk
    uint32_t* qr = q + (size_t)row * (K / 4);
    for (int i = lane * 4; i < k2; i += WAVE * 4) {
        uint4 w = *reinterpret_cast<const uint4*>(xr + i);
        uint2 o;
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            unsigned int packed = 0;
            packed = f32x2_to_fp8<false>(bf16_lo((&w.x)[h * 2]) * inv,
                                         bf16_hi((&w.x)[h * 2]) * inv, packed);
            packed = f32x2_to_fp8<true>(bf16_lo((&w.x)[h * 2 + 1]) * inv,
                                        bf16_hi((&w.x)[h * 2 + 1]) * inv, packed);
            (&o.x)[h] = packed;
        }
        *reinterpret_cast<uint2*>(qr + i / 2) = o;
    }
}

extern "C" int oa_quant_fp8(void* stream, const void* x, void* q, void* scales,
                            int T, int K) {
    if (K % 8 != 0) return -100;
    hipLaunchKernelGGL(quant_fp8_kernel, dim3(CEIL_DIV(T, 4)), dim3(256), 0,
                       (hipStream_t)stream, (const uint32_t*)x, (uint32_t*)q,
                       (float*)scales, T, K);
    HIP_CHECK_LAUNCH();
    return 0;
}

// ---- fp8-weight GEMV (decode path) -----------------------------------------
// out[M, N] = (x[M, K] @ W8[N, K]^T) * w_scale[n]; x bf16.
// NORM / ADDRES mirror the bf16 gemv fusions: rmsnorm prologue on x
// (fp32 rstd pass) and a residual-add epilogue.

// fp32 rstd of x row m (same as gemv.hip's row_rstd; separate TU)
__device__ __forceinline__ float row_rstd8(const uint32_t* xrow, int k2,
                                           int lane, float eps) {
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

template <int M, bool NORM, bool ADDRES>
__global__ __launch_bounds__(256) void gemv_fp8_kernel(
    const uint32_t* __restrict__ x,   // [M, K/2] bf16x2
    const uint32_t* __restrict__ w8,  // [N, K/4] fp8x4
    const float* __restrict__ wscale, // [N]
    uint32_t* __restrict__ out,       // [M, N] bf16
    const uint32_t* __restrict__ wn,  // [K/2] rmsnorm weight (NORM)
    const uint32_t* __restrict__ res, // [M, N] residual (ADDRES)
    int N, int K, float eps) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int k4 = K / 4;
    const int k2 = K / 2;

    float rstd[M];
    if (NORM) {
#pragma unroll
        for (int m = 0; m < M; ++m)
            rstd[m] = row_rstd8(x + (size_t)m * k2, k2, lane, eps);
    }

    for (int row = blockIdx.x * 4 + wid; row < N; row += gridDim.x * 4) {
        const uint32_t* wrow = w8 + (size_t)row * k4;
        float acc[M];
#pragma unroll
        for (int m = 0; m < M; ++m) acc[m] = 0.0f;
        // 16 B per lane = 16 fp8 elements per pass
        for (int i = lane * 4; i < k4; i += WAVE * 4) {
            u32x4 wv = nt_load4f(wrow + i);
            float wf[16];
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                f32x2 lo = fp8x2_to_f32<false>(wv[j]);
                f32x2 hi = fp8x2_to_f32<true>(wv[j]);
                wf[j * 4 + 0] = lo[0];
                wf[j * 4 + 1] = lo[1];
                wf[j * 4 + 2] = hi[0];
                wf[j * 4 + 3] = hi[1];
            }
#pragma unroll
            for (int m = 0; m < M; ++m) {
                // matching 16 bf16 of x = 2 x 16B loads
                uint4 xv0 = *reinterpret_cast<const uint4*>(x + (size_t)m * k2 + i * 2);
                uint4 xv1 = *reinterpret_cast<const uint4*>(x + (size_t)m * k2 + i * 2 + 4);
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    float x0l = bf16_lo((&xv0.x)[j]), x0h = bf16_hi((&xv0.x)[j]);
                    float x1l = bf16_lo((&xv1.x)[j]), x1h = bf16_hi((&xv1.x)[j]);
                    if (NORM) {
                        const uint32_t w0 = wn[i * 2 + j];
                        const uint32_t w1 = wn[i * 2 + 4 + j];
                        x0l *= rstd[m] * bf16_lo(w0);
                        x0h *= rstd[m] * bf16_hi(w0);
                        x1l *= rstd[m] * bf16_lo(w1);
                        x1h *= rstd[m] * bf16_hi(w1);
                    }
                    acc[m] = fmaf(x0l, wf[j * 2], acc[m]);
                    acc[m] = fmaf(x0h, wf[j * 2 + 1], acc[m]);
                    acc[m] = fmaf(x1l, wf[8 + j * 2], acc[m]);
                    acc[m] = fmaf(x1h, wf[8 + j * 2 + 1], acc[m]);
                }
            }
        }
        const float sc = wscale[row];
#pragma unroll
        for (int m = 0; m < M; ++m) {
            float v = wave_reduce_sum(acc[m]) * sc;
            if (lane == 0) {
                if (ADDRES)
                    v += bf16_to_f32(reinterpret_cast<const uint16_t*>(
                        res)[(size_t)m * N + row]);
                reinterpret_cast<uint16_t*>(out)[(size_t)m * N + row] =
                    f32_to_bf16(v);
            }
        }
    }
}

// Fused fp8 gate-up + SiLU: fused weight holds gate rows [0, I) and up rows
// [I, 2I); each wave streams BOTH fp8 rows and writes silu(g)*u directly.
template <int M, bool NORM>
__global__ __launch_bounds__(256) void gemv_gateup_fp8_kernel(
    const uint32_t* __restrict__ x, const uint32_t* __restrict__ w8,
    const float* __restrict__ wscale /* [2I] */, uint32_t* __restrict__ out,
    const uint32_t* __restrict__ wn, int I, int K, float eps) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int k4 = K / 4;
    const int k2 = K / 2;

    float rstd[M];
    if (NORM) {
#pragma unroll
        for (int m = 0; m < M; ++m)
            rstd[m] = row_rstd8(x + (size_t)m * k2, k2, lane, eps);
    }

    for (int row = blockIdx.x * 4 + wid; row < I; row += gridDim.x * 4) {
        const uint32_t* wg = w8 + (size_t)row * k4;
        const uint32_t* wu = w8 + (size_t)(row + I) * k4;
        float accg[M], accu[M];
#pragma unroll
        for (int m = 0; m < M; ++m) accg[m] = accu[m] = 0.0f;
        for (int i = lane * 4; i < k4; i += WAVE * 4) {
            u32x4 wvg = nt_load4f(wg + i);
            u32x4 wvu = nt_load4f(wu + i);
            float wfg[16], wfu[16];
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                f32x2 gl = fp8x2_to_f32<false>(wvg[j]);
                f32x2 gh = fp8x2_to_f32<true>(wvg[j]);
                f32x2 ul = fp8x2_to_f32<false>(wvu[j]);
                f32x2 uh = fp8x2_to_f32<true>(wvu[j]);
                wfg[j * 4 + 0] = gl[0]; wfg[j * 4 + 1] = gl[1];
                wfg[j * 4 + 2] = gh[0]; wfg[j * 4 + 3] = gh[1];
                wfu[j * 4 + 0] = ul[0]; wfu[j * 4 + 1] = ul[1];
                wfu[j * 4 + 2] = uh[0]; wfu[j * 4 + 3] = uh[1];
            }
#pragma unroll
            for (int m = 0; m < M; ++m) {
                uint4 xv0 = *reinterpret_cast<const uint4*>(x + (size_t)m * k2 + i * 2);
                uint4 xv1 = *reinterpret_cast<const uint4*>(x + (size_t)m * k2 + i * 2 + 4);
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    float x0l = bf16_lo((&xv0.x)[j]), x0h = bf16_hi((&xv0.x)[j]);
                    float x1l = bf16_lo((&xv1.x)[j]), x1h = bf16_hi((&xv1.x)[j]);
                    if (NORM) {
                        const uint32_t w0 = wn[i * 2 + j];
                        const uint32_t w1 = wn[i * 2 + 4 + j];
                        x0l *= rstd[m] * bf16_lo(w0);
                        x0h *= rstd[m] * bf16_hi(w0);
                        x1l *= rstd[m] * bf16_lo(w1);
                        x1h *= rstd[m] * bf16_hi(w1);
                    }
                    accg[m] = fmaf(x0l, wfg[j * 2], accg[m]);
                    accg[m] = fmaf(x0h, wfg[j * 2 + 1], accg[m]);
                    accg[m] = fmaf(x1l, wfg[8 + j * 2], accg[m]);
                    accg[m] = fmaf(x1h, wfg[8 + j * 2 + 1], accg[m]);
                    accu[m] = fmaf(x0l, wfu[j * 2], accu[m]);
                    accu[m] = fmaf(x0h, wfu[j * 2 + 1], accu[m]);
                    accu[m] = fmaf(x1l, wfu[8 + j * 2], accu[m]);
                    accu[m] = fmaf(x1h, wfu[8 + j * 2 + 1], accu[m]);
                }
            }
        }
        const float sg = wscale[row], su = wscale[row + I];
#pragma unroll
        for (int m = 0; m < M; ++m) {
            float g = wave_reduce_sum(accg[m]) * sg;
            float u = wave_reduce_sum(accu[m]) * su;
            if (lane == 0) {
                const float act = g / (1.0f + __expf(-g)) * u;
                reinterpret_cast<uint16_t*>(out)[(size_t)m * I + row] =
                    f32_to_bf16(act);
            }
        }
    }
}

// mode: 0 plain, 1 norm-prologue, 2 residual-add epilogue, 3 both
extern "C" int oa_gemv_fp8_ex(void* stream, const void* x, const void* w8,
                              const void* wscale, void* out, const void* wn,
                              const void* res, int M, int N, int K, float eps,
                              int mode) {
    if (K % 16 != 0) return -100;
    const int grid = min(2048, CEIL_DIV(N, 4));
#define LAUNCH_F8NM(MV, NORMV, RESV)                                           \
    hipLaunchKernelGGL((gemv_fp8_kernel<MV, NORMV, RESV>), dim3(grid),         \
                       dim3(256), 0, (hipStream_t)stream, (const uint32_t*)x,  \
                       (const uint32_t*)w8, (const float*)wscale,              \
                       (uint32_t*)out, (const uint32_t*)wn,                    \
                       (const uint32_t*)res, N, K, eps)
#define LAUNCH_F8(MV)                                                          \
    do {                                                                       \
        switch (mode) {                                                        \
            case 0: LAUNCH_F8NM(MV, false, false); break;                      \
            case 1: LAUNCH_F8NM(MV, true, false); break;                       \
            case 2: LAUNCH_F8NM(MV, false, true); break;                       \
            case 3: LAUNCH_F8NM(MV, true, true); break;                        \
            default: return -102;                                              \
        }                                                                      \
    } while (0)
    switch (M) {
        case 1: LAUNCH_F8(1); break;
        case 2: LAUNCH_F8(2); break;
        case 3: LAUNCH_F8(3); break;
        case 4: LAUNCH_F8(4); break;
        case 5: LAUNCH_F8(5); break;
        case 6: LAUNCH_F8(6); break;
        case 7: LAUNCH_F8(7); break;
        case 8: LAUNCH_F8(8); break;
        case 12: LAUNCH_F8(12); break;
        case 16: LAUNCH_F8(16); break;
        default: return -101;
    }
#undef LAUNCH_F8
#undef LAUNCH_F8NM
    HIP_CHECK_LAUNCH();
    return 0;
}

extern "C" int oa_gemv_fp8(void* stream, const void* x, const void* w8,
                           const void* wscale, void* out, int M, int N, int K) {
    return oa_gemv_fp8_ex(stream, x, w8, wscale, out, nullptr, nullptr, M, N,
                          K, 0.0f, 0);
}

extern "C" int oa_gemv_gateup_fp8(void* stream, const void* x, const void* w8,
                                  const void* wscale, void* out,
                                  const void* wn, int M, int I, int K,
                                  float eps, int norm) {
    if (K % 16 != 0) return -100;
    const int grid = min(2048, CEIL_DIV(I, 4));
#define LAUNCH_GU8(MV)                                                         \
    do {                                                                       \
        if (norm)                                                              \
            hipLaunchKernelGGL((gemv_gateup_fp8_kernel<MV, true>), dim3(grid), \
                               dim3(256), 0, (hipStream_t)stream,              \
                               (const uint32_t*)x, (const uint32_t*)w8,        \
                               (const float*)wscale, (uint32_t*)out,           \
                               (const uint32_t*)wn, I, K, eps);                \
        else                                                                   \
            hipLaunchKernelGGL((gemv_gateup_fp8_kernel<MV, false>),            \
                               dim3(grid), dim3(256), 0, (hipStream_t)stream,  \
                               (const uint32_t*)x, (const uint32_t*)w8,        \
                               (const float*)wscale, (uint32_t*)out,           \
                               (const uint32_t*)wn, I, K, eps);                \
    } while (0)
    switch (M) {
        case 1: LAUNCH_GU8(1); break;
        case 2: LAUNCH_GU8(2); break;
        case 3: LAUNCH_GU8(3); break;
        case 4: LAUNCH_GU8(4); break;
        case 6: LAUNCH_GU8(6); break;
        case 8: LAUNCH_GU8(8); break;
        default: return -101;
    }
#undef LAUNCH_GU8
    HIP_CHECK_LAUNCH();
    return 0;
}

// ---- fp8 MFMA tile GEMM (prefill path) -------------------------------------
// C[M, N] = (A8[M, K] @ B8[N, K]^T) * a_scale[m] * b_scale[n], bf16 out.
// 128x128 tile, 4 waves (2x2 of 64x64 per wave as 4x4 16x16 fragments),
// K-step 64, LDS-staged with the (row&15)<<4 XOR swizzle from the guide.
// v_mfma_f32_16x16x32_fp8_fp8: A lane l holds row l&15, k (l>>4)*8..+8
// (8 fp8 = 2 VGPRs); same C/D layout as the bf16 form (dtype-independent).

typedef __attribute__((ext_vector_type(2))) int fp8_frag;  // 8 fp8

#define GT 128   // tile M = N
#define GK 64    // K step (bytes per LDS row of fp8)
#define GKP 80   // padded LDS row stride: (row*80/4)%64 = row*20 mod 64 is
                 // distinct for 16 consecutive rows -> conflict-free
                 // fragment reads (guide Guideline 4 pad-by-access-width)

__device__ __forceinline__ uint32_t a8_swz(int row, int byte_in_row) {
    return (uint32_t)(row * GKP + byte_in_row);
}

__global__ __launch_bounds__(256, 2) void gemm_fp8_kernel(
    const uint32_t* __restrict__ a8,  // [M, K/4]
    const uint32_t* __restrict__ b8,  // [N, K/4]
    const float* __restrict__ ascale, // [M]
    const float* __restrict__ bscale, // [N]
    uint32_t* __restrict__ c,         // [M, N/2] bf16x2
    int M, int N, int K) {
    __shared__ __attribute__((aligned(16))) char smem[2 * GT * GKP];  // A then B

    const int tm = blockIdx.x * GT;
    const int tn = blockIdx.y * GT;
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int wr = wid >> 1;   // wave row 0..1 (owns 64 rows)
    const int wc = wid & 1;    // wave col 0..1 (owns 64 cols)
    const int fr = lane & 15;
    const int fs = lane >> 4;

    f32x4_t acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4_t){0.f, 0.f, 0.f, 0.f};

    const int k4 = K / 4;
    for (int k0 = 0; k0 < K; k0 += GK) {
        __syncthreads();
        // stage A and B tiles: 128 rows x 64 fp8 = 8 KiB each; 256 threads
        // x 16 B x 2 pieces per operand
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            const int li = threadIdx.x + p * 256;   // 0..511
            const int row = li >> 2;                // 0..127
            const int b16 = (li & 3) * 16;          // byte in row (of 64)
            uint4 av = make_uint4(0, 0, 0, 0), bv = make_uint4(0, 0, 0, 0);
            if (tm + row < M)
                av = *reinterpret_cast<const uint4*>(a8 + (size_t)(tm + row) * k4 + (k0 + b16) / 4);
            if (tn + row < N)
                bv = *reinterpret_cast<const uint4*>(b8 + (size_t)(tn + row) * k4 + (k0 + b16) / 4);
            *reinterpret_cast<uint4*>(smem + a8_swz(row, b16)) = av;
            *reinterpret_cast<uint4*>(smem + GT * GKP + a8_swz(row, b16)) = bv;
        }
        __syncthreads();

        // 4x4 fragment tiles x (GK/32 = 2) k-steps
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            const int kb = kk * 32 + fs * 8;  // this lane's 8 fp8 within the k-step
            fp8_frag afrag[4], bfrag[4];
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int arow = wr * 64 + i * 16 + fr;
                afrag[i] = *reinterpret_cast<const fp8_frag*>(smem + a8_swz(arow, kb));
                const int brow = wc * 64 + i * 16 + fr;
                bfrag[i] = *reinterpret_cast<const fp8_frag*>(smem + GT * GKP + a8_swz(brow, kb));
            }
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
                        *reinterpret_cast<long*>(&afrag[i]),
                        *reinterpret_cast<long*>(&bfrag[j]), acc[i][j], 0, 0, 0);
        }
    }

    // epilogue: C[row][col] = acc * ascale[row] * bscale[col]
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = tm + wr * 64 + i * 16 + fs * 4 + r;
                const int col = tn + wc * 64 + j * 16 + fr;
                if (row < M && col < N) {
                    const float v = acc[i][j][r] * ascale[row] * bscale[col];
                    reinterpret_cast<uint16_t*>(c)[(size_t)row * N + col] = f32_to_bf16(v);
                }
            }
        }
    }
}

extern "C" int oa_gemm_fp8(void* stream, const void* a8, const void* b8,
                           const void* ascale, const void* bscale, void* c,
                           int M, int N, int K) {
    if (K % GK != 0) return -100;
    dim3 grid(CEIL_DIV(M, GT), CEIL_DIV(N, GT)), block(256);
    hipLaunchKernelGGL(gemm_fp8_kernel, grid, block, 0, (hipStream_t)stream,
                       (const uint32_t*)a8, (const uint32_t*)b8,
                       (const float*)ascale, (const float*)bscale, (uint32_t*)c,
                       M, N, K);
    HIP_CHECK_LAUNCH();
    return 0;
}
