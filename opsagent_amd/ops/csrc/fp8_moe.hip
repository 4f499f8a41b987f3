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

// XS: stage x as PRE-SCALED f32 in LDS once per block (requires M*K*4 <=
// 128 KiB). The fp8 dequant stream loop was VALU-bound at ~4.5 TB/s: per
// 16 W bytes it spent ~16 VALU unpacking the bf16 x (plus 4 more per
// element on the NORM path); the f32 image folds the rmsnorm weight and
// rstd in at fill time, leaving cvt+fma only in the stream loop.
// RW=2: two adjacent W rows per wave — the (expensive) bf16 x unpack
// amortizes over both fp8 streams, the same structure that makes the
// gateup form the fastest fp8 VALU stream (5.96 vs 5.1 TB/s measured).
template <int M, bool NORM, bool ADDRES, bool XS = false, int RW = 1>
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

    extern __shared__ float xs[];  // [M][K] f32, XS only
    if (XS) {
        for (int idx = threadIdx.x; idx < M * k2; idx += blockDim.x) {
            const int m = idx / k2;
            const int i2 = idx % k2;  // word index in row
            const uint32_t xv = x[(size_t)m * k2 + i2];
            float lo = bf16_lo(xv), hi = bf16_hi(xv);
            if (NORM) {
                const uint32_t w = wn[i2];
                lo *= rstd[m] * bf16_lo(w);
                hi *= rstd[m] * bf16_hi(w);
            }
            xs[(size_t)m * K + i2 * 2] = lo;
            xs[(size_t)m * K + i2 * 2 + 1] = hi;
        }
        __syncthreads();
    }

    for (int row0 = (blockIdx.x * 4 + wid) * RW; row0 < N;
         row0 += gridDim.x * 4 * RW) {
        const uint32_t* wrow[RW];
        bool live[RW];
#pragma unroll
        for (int rr = 0; rr < RW; ++rr) {
            live[rr] = row0 + rr < N;
            wrow[rr] = w8 + (size_t)(live[rr] ? row0 + rr : row0) * k4;
        }
        float acc[RW][M];
#pragma unroll
        for (int rr = 0; rr < RW; ++rr)
#pragma unroll
            for (int m = 0; m < M; ++m) acc[rr][m] = 0.0f;
        // 16 B per lane = 16 fp8 elements per pass per row stream; the
        // bf16 x unpack happens ONCE and feeds every stream
        for (int i = lane * 4; i < k4; i += WAVE * 4) {
            float wf[RW][16];
#pragma unroll
            for (int rr = 0; rr < RW; ++rr) {
                u32x4 wv = nt_load4f(wrow[rr] + i);
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    f32x2 lo = fp8x2_to_f32<false>(wv[j]);
                    f32x2 hi = fp8x2_to_f32<true>(wv[j]);
                    wf[rr][j * 4 + 0] = lo[0];
                    wf[rr][j * 4 + 1] = lo[1];
                    wf[rr][j * 4 + 2] = hi[0];
                    wf[rr][j * 4 + 3] = hi[1];
                }
            }
#pragma unroll
            for (int m = 0; m < M; ++m) {
                float xf[16];
                if (XS) {
                    const float* xr = xs + (size_t)m * K + i * 4;
#pragma unroll
                    for (int e = 0; e < 16; ++e) xf[e] = xr[e];
                } else {
                    // matching 16 bf16 of x = 2 x 16B loads
                    uint4 xv0 = *reinterpret_cast<const uint4*>(
                        x + (size_t)m * k2 + i * 2);
                    uint4 xv1 = *reinterpret_cast<const uint4*>(
                        x + (size_t)m * k2 + i * 2 + 4);
#pragma unroll
                    for (int j = 0; j < 4; ++j) {
                        float x0l = bf16_lo((&xv0.x)[j]),
                              x0h = bf16_hi((&xv0.x)[j]);
                        float x1l = bf16_lo((&xv1.x)[j]),
                              x1h = bf16_hi((&xv1.x)[j]);
                        if (NORM) {
                            const uint32_t w0 = wn[i * 2 + j];
                            const uint32_t w1 = wn[i * 2 + 4 + j];
                            x0l *= rstd[m] * bf16_lo(w0);
                            x0h *= rstd[m] * bf16_hi(w0);
                            x1l *= rstd[m] * bf16_lo(w1);
                            x1h *= rstd[m] * bf16_hi(w1);
                        }
                        xf[j * 2] = x0l;
                        xf[j * 2 + 1] = x0h;
                        xf[8 + j * 2] = x1l;
                        xf[8 + j * 2 + 1] = x1h;
                    }
                }
#pragma unroll
                for (int rr = 0; rr < RW; ++rr)
#pragma unroll
                    for (int e = 0; e < 16; ++e)
                        acc[rr][m] = fmaf(xf[e], wf[rr][e], acc[rr][m]);
            }
        }
#pragma unroll
        for (int rr = 0; rr < RW; ++rr) {
            const float sc = wscale[live[rr] ? row0 + rr : row0];
#pragma unroll
            for (int m = 0; m < M; ++m) {
                float v = wave_reduce_sum(acc[rr][m]) * sc;
                if (lane == 0 && live[rr]) {
                    if (ADDRES)
                        v += bf16_to_f32(reinterpret_cast<const uint16_t*>(
                            res)[(size_t)m * N + row0 + rr]);
                    reinterpret_cast<uint16_t*>(
                        out)[(size_t)m * N + row0 + rr] = f32_to_bf16(v);
                }
            }
        }
    }
}

// Fused fp8 gate-up + SiLU: fused weight holds gate rows [0, I) and up rows
// [I, 2I); each wave streams BOTH fp8 rows and writes silu(g)*u directly.
template <int M, bool NORM, bool XS = false>
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

    extern __shared__ float xs[];  // [M][K] f32, XS only (pre-scaled)
    if (XS) {
        for (int idx = threadIdx.x; idx < M * k2; idx += blockDim.x) {
            const int m = idx / k2;
            const int i2 = idx % k2;
            const uint32_t xv = x[(size_t)m * k2 + i2];
            float lo = bf16_lo(xv), hi = bf16_hi(xv);
            if (NORM) {
                const uint32_t w = wn[i2];
                lo *= rstd[m] * bf16_lo(w);
                hi *= rstd[m] * bf16_hi(w);
            }
            xs[(size_t)m * K + i2 * 2] = lo;
            xs[(size_t)m * K + i2 * 2 + 1] = hi;
        }
        __syncthreads();
    }

    // RW adjacent row-pairs per wave at M=1: 2*RW fp8 streams share ONE x
    // unpack per pass (the amortization that took the plain kernel to RW=2)
    constexpr int RW = (M == 1 && !XS) ? 2 : 1;
    for (int row0 = (blockIdx.x * 4 + wid) * RW; row0 < I;
         row0 += gridDim.x * 4 * RW) {
        bool liver[RW];
        const uint32_t* wgr[RW];
        const uint32_t* wur[RW];
#pragma unroll
        for (int rr = 0; rr < RW; ++rr) {
            liver[rr] = row0 + rr < I;
            const int r = liver[rr] ? row0 + rr : row0;
            wgr[rr] = w8 + (size_t)r * k4;
            wur[rr] = w8 + (size_t)(r + I) * k4;
        }
        float accg[RW][M], accu[RW][M];
#pragma unroll
        for (int rr = 0; rr < RW; ++rr)
#pragma unroll
            for (int m = 0; m < M; ++m) accg[rr][m] = accu[rr][m] = 0.0f;
        for (int i = lane * 4; i < k4; i += WAVE * 4) {
          if constexpr (RW == 2) {
            // M == 1 here: unpack x ONCE, feed all four fp8 streams
            uint4 xv0 = *reinterpret_cast<const uint4*>(x + i * 2);
            uint4 xv1 = *reinterpret_cast<const uint4*>(x + i * 2 + 4);
            float xf[16];
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                float x0l = bf16_lo((&xv0.x)[j]), x0h = bf16_hi((&xv0.x)[j]);
                float x1l = bf16_lo((&xv1.x)[j]), x1h = bf16_hi((&xv1.x)[j]);
                if (NORM) {
                    const uint32_t w0 = wn[i * 2 + j];
                    const uint32_t w1 = wn[i * 2 + 4 + j];
                    x0l *= rstd[0] * bf16_lo(w0);
                    x0h *= rstd[0] * bf16_hi(w0);
                    x1l *= rstd[0] * bf16_lo(w1);
                    x1h *= rstd[0] * bf16_hi(w1);
                }
                xf[j * 2] = x0l;
                xf[j * 2 + 1] = x0h;
                xf[8 + j * 2] = x1l;
                xf[8 + j * 2 + 1] = x1h;
            }
#pragma unroll
            for (int rr = 0; rr < RW; ++rr) {
                u32x4 wvg = nt_load4f(wgr[rr] + i);
                u32x4 wvu = nt_load4f(wur[rr] + i);
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    f32x2 gl = fp8x2_to_f32<false>(wvg[j]);
                    f32x2 gh = fp8x2_to_f32<true>(wvg[j]);
                    f32x2 ul = fp8x2_to_f32<false>(wvu[j]);
                    f32x2 uh = fp8x2_to_f32<true>(wvu[j]);
                    accg[rr][0] = fmaf(xf[j * 4 + 0], gl[0], accg[rr][0]);
                    accg[rr][0] = fmaf(xf[j * 4 + 1], gl[1], accg[rr][0]);
                    accg[rr][0] = fmaf(xf[j * 4 + 2], gh[0], accg[rr][0]);
                    accg[rr][0] = fmaf(xf[j * 4 + 3], gh[1], accg[rr][0]);
                    accu[rr][0] = fmaf(xf[j * 4 + 0], ul[0], accu[rr][0]);
                    accu[rr][0] = fmaf(xf[j * 4 + 1], ul[1], accu[rr][0]);
                    accu[rr][0] = fmaf(xf[j * 4 + 2], uh[0], accu[rr][0]);
                    accu[rr][0] = fmaf(xf[j * 4 + 3], uh[1], accu[rr][0]);
                }
            }
            continue;
          }

            u32x4 wvg = nt_load4f(wgr[0] + i);
            u32x4 wvu = nt_load4f(wur[0] + i);
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
                if (XS) {
                    const float* xr = xs + (size_t)m * K + i * 4;
#pragma unroll
                    for (int q = 0; q < 4; ++q) {
                        float4 a = *reinterpret_cast<const float4*>(xr + q * 4);
                        accg[0][m] = fmaf(a.x, wfg[q * 4 + 0], accg[0][m]);
                        accg[0][m] = fmaf(a.y, wfg[q * 4 + 1], accg[0][m]);
                        accg[0][m] = fmaf(a.z, wfg[q * 4 + 2], accg[0][m]);
                        accg[0][m] = fmaf(a.w, wfg[q * 4 + 3], accg[0][m]);
                        accu[0][m] = fmaf(a.x, wfu[q * 4 + 0], accu[0][m]);
                        accu[0][m] = fmaf(a.y, wfu[q * 4 + 1], accu[0][m]);
                        accu[0][m] = fmaf(a.z, wfu[q * 4 + 2], accu[0][m]);
                        accu[0][m] = fmaf(a.w, wfu[q * 4 + 3], accu[0][m]);
                    }
                    continue;
                }
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
                    accg[0][m] = fmaf(x0l, wfg[j * 2], accg[0][m]);
                    accg[0][m] = fmaf(x0h, wfg[j * 2 + 1], accg[0][m]);
                    accg[0][m] = fmaf(x1l, wfg[8 + j * 2], accg[0][m]);
                    accg[0][m] = fmaf(x1h, wfg[8 + j * 2 + 1], accg[0][m]);
                    accu[0][m] = fmaf(x0l, wfu[j * 2], accu[0][m]);
                    accu[0][m] = fmaf(x0h, wfu[j * 2 + 1], accu[0][m]);
                    accu[0][m] = fmaf(x1l, wfu[8 + j * 2], accu[0][m]);
                    accu[0][m] = fmaf(x1h, wfu[8 + j * 2 + 1], accu[0][m]);
                }
            }
        }
#pragma unroll
        for (int rr = 0; rr < RW; ++rr) {
            const int row = liver[rr] ? row0 + rr : row0;
            const float sg = wscale[row], su = wscale[row + I];
#pragma unroll
            for (int m = 0; m < M; ++m) {
                float g = wave_reduce_sum(accg[rr][m]) * sg;
                float u = wave_reduce_sum(accu[rr][m]) * su;
                if (lane == 0 && liver[rr]) {
                    const float act = g / (1.0f + __expf(-g)) * u;
                    reinterpret_cast<uint16_t*>(out)[(size_t)m * I + row] =
                        f32_to_bf16(act);
                }
            }
        }
    }
}

// mode: 0 plain, 1 norm-prologue, 2 residual-add epilogue, 3 both
extern "C" int oa_gemv_fp8_mfma(void* stream, const void* x, const void* w8,
                                const void* wscale, void* out, const void* wn,
                                const void* res, int M, int N, int K,
                                float eps, int mode, int gateup);

// MFMA path eligibility (see the v3 kernel at the end of this file).
// A/B-measured (profiles/README.md, scripts/gemv_fp8_bench.py): the MFMA
// stream wins for M >= 2 (batched decode: +45..+59% at M=4 — the extra
// batch columns ride the same weight stream for free) while the VALU
// kernel's contiguous row-per-wave streams stay ahead at M=1 (5.1 vs 3.1
// TB/s: 16 strided row streams per wave + the x-quant launch don't pay off
// for one column). OPSAGENT_FP8_GEMV_MFMA=0 forces VALU, =2 forces MFMA.
static bool gemv_fp8_use_mfma(int M, int rows, int K, int gateup) {
    const char* e = getenv("OPSAGENT_FP8_GEMV_MFMA");
    if (e && e[0] == '0') return false;
    // must mirror oa_gemv_fp8_mfma's guards exactly (PF-deep prefetch ring
    // needs (K/512) % PF == 0) so ineligible shapes FALL BACK, not error
    if (K % 512 != 0 || (K / 512) % (gateup ? 4 : 8) != 0 || K > 32768 ||
        M > 16)
        return false;
    if (M < 2 && !(e && e[0] == '2')) return false;
    (void)rows;
    return (size_t)M * K <= 128 * 1024;
}

extern "C" int oa_gemv_fp8_ex(void* stream, const void* x, const void* w8,
                              const void* wscale, void* out, const void* wn,
                              const void* res, int M, int N, int K, float eps,
                              int mode) {
    if (K % 16 != 0) return -100;
    if (gemv_fp8_use_mfma(M, N, K, 0))
        return oa_gemv_fp8_mfma(stream, x, w8, wscale, out, wn, res, M, N, K,
                                eps, mode, 0);
    // M=1 runs the RW=2 two-stream form (x unpack amortized — the gateup
    // structure measured 5.96 vs 5.1 TB/s); OPSAGENT_FP8_GEMV_RW=1 forces
    // the single-stream form for A/B.
    const char* rwe = getenv("OPSAGENT_FP8_GEMV_RW");
    const int rw = (M == 1 && !(rwe && rwe[0] == '1')) ? 2 : 1;
    const int grid = min(4096, CEIL_DIV(N, 4 * rw));
    // pre-scaled f32 x image in LDS: measured NET NEGATIVE on the 70B fp8
    // turn (747 vs 503 ms — the 32-114 KiB LDS footprint collapses block
    // occupancy, which this latency-hiding structure needs more than the
    // saved unpack VALU). Default OFF; OPSAGENT_FP8_GEMV_XS=1 re-enables
    // for A/B.
    const size_t xs_bytes = (size_t)M * K * 4;
    const char* xse = getenv("OPSAGENT_FP8_GEMV_XS");
    const bool use_xs = xs_bytes <= 131072 && xse && xse[0] == '1';
#define LAUNCH_F8NM(MV, NORMV, RESV)                                           \
    do {                                                                       \
        if (use_xs)                                                            \
            hipLaunchKernelGGL((gemv_fp8_kernel<MV, NORMV, RESV, true>),       \
                               dim3(grid), dim3(256), xs_bytes,                \
                               (hipStream_t)stream, (const uint32_t*)x,        \
                               (const uint32_t*)w8, (const float*)wscale,      \
                               (uint32_t*)out, (const uint32_t*)wn,            \
                               (const uint32_t*)res, N, K, eps);               \
        else if (rw == 2)                                                      \
            hipLaunchKernelGGL(                                                \
                (gemv_fp8_kernel<MV, NORMV, RESV, false, 2>), dim3(grid),      \
                dim3(256), 0, (hipStream_t)stream, (const uint32_t*)x,         \
                (const uint32_t*)w8, (const float*)wscale, (uint32_t*)out,     \
                (const uint32_t*)wn, (const uint32_t*)res, N, K, eps);         \
        else                                                                   \
            hipLaunchKernelGGL((gemv_fp8_kernel<MV, NORMV, RESV, false>),      \
                               dim3(grid), dim3(256), 0, (hipStream_t)stream,  \
                               (const uint32_t*)x, (const uint32_t*)w8,        \
                               (const float*)wscale, (uint32_t*)out,           \
                               (const uint32_t*)wn, (const uint32_t*)res, N,   \
                               K, eps);                                        \
    } while (0)
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
    if (gemv_fp8_use_mfma(M, I, K, 1))
        return oa_gemv_fp8_mfma(stream, x, w8, wscale, out, wn, nullptr, M,
                                2 * I, K, eps, norm ? 1 : 0, 1);
    // M=1 instantiates RW=2 row-pairs (4 streams/wave)
    const int grid = min(4096, CEIL_DIV(I, M == 1 ? 8 : 4));
    const size_t xs_bytes = (size_t)M * K * 4;
    const char* xse = getenv("OPSAGENT_FP8_GEMV_XS");
    const bool use_xs = xs_bytes <= 131072 && xse && xse[0] == '1';
#define LAUNCH_GU8_1(MV, NORMV, XSV, SH)                                       \
    hipLaunchKernelGGL((gemv_gateup_fp8_kernel<MV, NORMV, XSV>), dim3(grid),   \
                       dim3(256), SH, (hipStream_t)stream,                     \
                       (const uint32_t*)x, (const uint32_t*)w8,                \
                       (const float*)wscale, (uint32_t*)out,                   \
                       (const uint32_t*)wn, I, K, eps)
#define LAUNCH_GU8(MV)                                                         \
    do {                                                                       \
        if (norm) {                                                            \
            if (use_xs) LAUNCH_GU8_1(MV, true, true, xs_bytes);                \
            else LAUNCH_GU8_1(MV, true, false, 0);                             \
        } else {                                                               \
            if (use_xs) LAUNCH_GU8_1(MV, false, true, xs_bytes);               \
            else LAUNCH_GU8_1(MV, false, false, 0);                            \
        }                                                                      \
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
#undef LAUNCH_GU8_1
    HIP_CHECK_LAUNCH();
    return 0;
}

// ---- fp8 MFMA tile GEMM (prefill path) -------------------------------------
// C[M, N] = (A8[M, K] @ B8[N, K]^T) * a_scale[m] * b_scale[n], bf16 out.
//
// v2 (VERDICT r1 #3 — pipeline the fp8 GEMM): 128x128 tile, 4 waves (2x2 of
// 64x64), K-step 128 fp8 bytes (2x the old unroll -> half the barriers), A/B
// staged by 16-B global_load_lds into a DOUBLE-BUFFERED ring (the guide's
// step-3 '+67%' lever over register staging, and its verified 2-buffer
// "glds + vmcnt(0) + __syncthreads" form). The XOR swizzle moves to the DMA
// source address (glds writes lane-linear): byte ^= (row&7)<<4 spreads the
// 16-lane fragment-read groups over the banks at 16-B granularity.
// v_mfma_f32_16x16x32_fp8_fp8: A lane l holds row l&15, k (l>>4)*8..+8
// (8 fp8 = 2 VGPRs); same C/D layout as the bf16 form (dtype-independent).
// Tail rows (M or N not multiple of 128) CLAMP the DMA source row and the
// epilogue masks the store — garbage products are never written.

typedef __attribute__((ext_vector_type(2))) int fp8_frag;   // 8 fp8
typedef __attribute__((ext_vector_type(8))) int i32x8_f8;   // 32 fp8 (MX operand)
typedef __attribute__((ext_vector_type(16))) float f32x16_t;  // 32x32 C/D

#define GT 128    // tile M = N
#define GK2 128   // K step in fp8 bytes
#define GTILE_B (GT * GK2)  // 16 KiB per operand per buffer

__device__ __forceinline__ uint32_t g8_swz(int row, int byte_in_row) {
    return (uint32_t)(row * GK2 + (byte_in_row ^ ((row & 7) << 4)));
}

// GLDS=true: LDS-DMA staging (fastest when the operands are L3-resident).
// GLDS=false: register staging (plain loads -> regs under the MFMAs ->
// ds_write after the barrier, T14 form): the LDS-DMA gather path collapses
// to ~0.9 TB/s chip-wide when the streamed operand MISSES the L3 (measured
// 119 TF at M1024 N8192 K28672 where B alone is 235 MB), while plain loads
// stream HBM at full rate.
template <bool GLDS>
__global__ __launch_bounds__(256, 1) void gemm_fp8_kernel_v2(
    const uint32_t* __restrict__ a8,  // [M, K/4]
    const uint32_t* __restrict__ b8,  // [N, K/4]
    const float* __restrict__ ascale, // [M]
    const float* __restrict__ bscale, // [N]
    uint32_t* __restrict__ c,         // [M, N/2] bf16x2
    int M, int N, int K) {
    // ring layout: [buf][A|B] 16 KiB quadrants
    __shared__ __attribute__((aligned(16))) char smem[4 * GTILE_B];

    // N-tiles on x: the dispatcher places block b on XCD b%8, so an M-major
    // x axis pins each M-tile's 224-block column to ONE XCD (measured 9x
    // slowdown at N=28672 where B exceeds the L3); N-major spreads the
    // B stream across XCDs and co-resident blocks share the small A tiles.
    const int tm = blockIdx.y * GT;
    const int tn = blockIdx.x * GT;
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int wr = wid >> 1;   // wave row 0..1 (owns 64 rows)
    const int wc = wid & 1;    // wave col 0..1 (owns 64 cols)
    const int fr = lane & 15;
    const int fs = lane >> 4;

    f32x16_t accw[2][2] = {};

    // staging: per K-step each operand moves 16 KiB = 1024 16-B pieces;
    // 4 waves x 4 chunks x 64 lanes. Chunk c covers rows c*8..c*8+7
    // (8 rows x 128 B); lane l -> row c*8 + l/8, byte (l%8)*16 ^ swz.
    // Same asm-glds + persistent-operand discipline as attention_prefill
    // (see the comment there: the builtin form's DMA gets drained at the
    // first ds_read of every K-step by the wait inserter).
    const int arow_l = lane >> 3;           // row within chunk
    const int abyte_l = ((lane & 7) * 16) ^ ((arow_l & 7) << 4);
    const char* abase = reinterpret_cast<const char*>(a8);
    const char* bbase = reinterpret_cast<const char*>(b8);

    uint32_t m0a[2][4], m0b[2][4];
    if constexpr (GLDS) {
#pragma unroll
        for (int buf = 0; buf < 2; ++buf)
#pragma unroll
            for (int p = 0; p < 4; ++p) {
                const int ch = wid * 4 + p;  // chunk 0..15
                m0a[buf][p] = __builtin_amdgcn_readfirstlane(
                    (uint32_t)(uintptr_t)smem + buf * 2 * GTILE_B + ch * 1024);
                m0b[buf][p] = __builtin_amdgcn_readfirstlane(
                    (uint32_t)(uintptr_t)smem + buf * 2 * GTILE_B + GTILE_B +
                    ch * 1024);
            }
    }
    const char* asrc[4];
    const char* bsrc[4];
    uint32_t m0a_cur[4], m0b_cur[4];
    auto set_step = [&](int k0, int buf) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            const int ch = wid * 4 + p;
            const int ar = min(tm + ch * 8 + arow_l, M - 1);
            const int br = min(tn + ch * 8 + arow_l, N - 1);
            asrc[p] = abase + (size_t)ar * (K) + k0 + abyte_l;
            bsrc[p] = bbase + (size_t)br * (K) + k0 + abyte_l;
            m0a_cur[p] = m0a[buf][p];
            m0b_cur[p] = m0b[buf][p];
        }
    };
    auto stage = [&]() {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            asm volatile(
                "s_mov_b32 m0, %0\n\t"
                "global_load_lds_dwordx4 %1, off"
                :
                : "s"(m0a_cur[p]), "v"(asrc[p]));
            asm volatile(
                "s_mov_b32 m0, %0\n\t"
                "global_load_lds_dwordx4 %1, off"
                :
                : "s"(m0b_cur[p]), "v"(bsrc[p]));
        }
    };
    auto drain = [&]() {
        asm volatile("s_waitcnt vmcnt(0)"
                     :
                     : "v"(asrc[0]), "v"(asrc[1]), "v"(asrc[2]), "v"(asrc[3]),
                       "v"(bsrc[0]), "v"(bsrc[1]), "v"(bsrc[2]), "v"(bsrc[3]),
                       "s"(m0a_cur[0]), "s"(m0a_cur[1]), "s"(m0a_cur[2]),
                       "s"(m0a_cur[3]), "s"(m0b_cur[0]), "s"(m0b_cur[1]),
                       "s"(m0b_cur[2]), "s"(m0b_cur[3])
                     : "memory");
    };

    // Inner product on v_mfma_scale_f32_32x32x64_f8f6f4 with IDENTITY E8M0
    // scales (0x7f = 2^0): the large-K scaled forms are the ONLY fp8 MFMAs
    // past the bf16 rate on gfx950 (~4.6 PF vs ~2.1 PF for 16x16x32 fp8 —
    // guide §3); with unit scales the numerics are exactly the plain-fp8
    // kernel's (fp8 products, f32 accumulation; per-row float scales stay
    // in the epilogue). Fragment maps: A lane l&31 = row, (l>>5) = 32-byte
    // k-half; B symmetric over columns; C/D is the standard 32x32 map.
    auto compute = [&](const char* abuf, const char* bbuf) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {          // two K=64 steps per tile
            const int kb = kk * 64 + (lane >> 5) * 32;  // this lane's 32 B
            i32x8_f8 afrag[2], bfrag[2];
#pragma unroll
            for (int i = 0; i < 2; ++i) {
                const int arow = wr * 64 + i * 32 + (lane & 31);
                const int brow = wc * 64 + i * 32 + (lane & 31);
                // 32 contiguous k-bytes under the 16-B-granular XOR swizzle
                // = two independent 16-B reads
                *reinterpret_cast<uint4*>(&afrag[i]) =
                    *reinterpret_cast<const uint4*>(abuf + g8_swz(arow, kb));
                *(reinterpret_cast<uint4*>(&afrag[i]) + 1) =
                    *reinterpret_cast<const uint4*>(abuf + g8_swz(arow, kb + 16));
                *reinterpret_cast<uint4*>(&bfrag[i]) =
                    *reinterpret_cast<const uint4*>(bbuf + g8_swz(brow, kb));
                *(reinterpret_cast<uint4*>(&bfrag[i]) + 1) =
                    *reinterpret_cast<const uint4*>(bbuf + g8_swz(brow, kb + 16));
            }
#pragma unroll
            for (int i = 0; i < 2; ++i)
#pragma unroll
                for (int j = 0; j < 2; ++j)
                    accw[i][j] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
                        afrag[i], bfrag[j], accw[i][j],
                        0 /*cbsz: fp8*/, 0 /*blgp: fp8*/,
                        0, 0x7f, 0, 0x7f);
        }
    };

    // register-staging state (GLDS=false): one set, T14 write-after-barrier
    uint4 areg[4], breg[4];
    auto load_step = [&](int k0) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            const int ch = wid * 4 + p;
            const int ar = min(tm + ch * 8 + arow_l, M - 1);
            const int br = min(tn + ch * 8 + arow_l, N - 1);
            areg[p] = *reinterpret_cast<const uint4*>(
                abase + (size_t)ar * K + k0 + abyte_l);
            breg[p] = *reinterpret_cast<const uint4*>(
                bbase + (size_t)br * K + k0 + abyte_l);
        }
    };
    auto write_step = [&](int buf) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            const int ch = wid * 4 + p;
            char* dst = smem + buf * 2 * GTILE_B + ch * 1024 + lane * 16;
            *reinterpret_cast<uint4*>(dst) = areg[p];
            *reinterpret_cast<uint4*>(dst + GTILE_B) = breg[p];
        }
    };

    const int nsteps = K / GK2;
    if constexpr (GLDS) {
        set_step(0, 0);
        stage();
        drain();
        __syncthreads();
        for (int t = 0; t < nsteps; ++t) {
            const int cur = t & 1;
            if (t + 1 < nsteps) {
                set_step((t + 1) * GK2, cur ^ 1);
                stage();
            }
            compute(smem + cur * 2 * GTILE_B, smem + cur * 2 * GTILE_B + GTILE_B);
            drain();
            __syncthreads();
        }
    } else {
        load_step(0);
        write_step(0);
        __syncthreads();
        for (int t = 0; t < nsteps; ++t) {
            const int cur = t & 1;
            if (t + 1 < nsteps) load_step((t + 1) * GK2);  // HBM hides under MFMAs
            compute(smem + cur * 2 * GTILE_B, smem + cur * 2 * GTILE_B + GTILE_B);
            if (t + 1 < nsteps) write_step(cur ^ 1);       // other buffer: no wait on readers
            __syncthreads();
        }
    }

    // epilogue: C[row][col] = acc * ascale[row] * bscale[col]
    // 32x32 C/D map: col = lane&31, row = (r&3) + 8*(r>>2) + 4*(lane>>5)
#pragma unroll
    for (int i = 0; i < 2; ++i) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = tm + wr * 64 + i * 32 + (r & 3) + 8 * (r >> 2) +
                                4 * (lane >> 5);
                const int col = tn + wc * 64 + j * 32 + (lane & 31);
                if (row < M && col < N) {
                    const float v = accw[i][j][r] * ascale[row] * bscale[col];
                    reinterpret_cast<uint16_t*>(c)[(size_t)row * N + col] = f32_to_bf16(v);
                }
            }
        }
    }
}

extern "C" int oa_gemm_fp8(void* stream, const void* a8, const void* b8,
                           const void* ascale, const void* bscale, void* c,
                           int M, int N, int K) {
    if (K % GK2 != 0) return -100;
    dim3 grid(CEIL_DIV(N, GT), CEIL_DIV(M, GT)), block(256);
    // Staging choice (3-run A/B on MI355X): glds staging wins at every
    // shape (1083-1551 TF vs registers' 541-764) EXCEPT deep-K (K=28672:
    // reproducible 119-129 TF collapse of the DMA gather — both operands
    // at 28 KiB row stride with the B panel re-read per M-tile; register
    // staging holds 735-764 TF there). A separate SPORADIC ~4.2 ms mode
    // was observed hitting hipBLASLt itself on the same boxes — box
    // throttling, not kernel-attributable. OPSAGENT_FP8_STAGE=glds|reg
    // forces a path (A/B).
    const char* fs_env = getenv("OPSAGENT_FP8_STAGE");
    bool use_glds = K <= 16384;
    if (fs_env && fs_env[0] == 'g') use_glds = true;
    if (fs_env && fs_env[0] == 'r') use_glds = false;
    if (use_glds)
        hipLaunchKernelGGL((gemm_fp8_kernel_v2<true>), grid, block, 0,
                           (hipStream_t)stream, (const uint32_t*)a8,
                           (const uint32_t*)b8, (const float*)ascale,
                           (const float*)bscale, (uint32_t*)c, M, N, K);
    else
        hipLaunchKernelGGL((gemm_fp8_kernel_v2<false>), grid, block, 0,
                           (hipStream_t)stream, (const uint32_t*)a8,
                           (const uint32_t*)b8, (const float*)ascale,
                           (const float*)bscale, (uint32_t*)c, M, N, K);
    HIP_CHECK_LAUNCH();
    return 0;
}


// ---- MFMA fp8 GEMV v2 (decode path) ----------------------------------------
// The VALU dequant stream tops out at ~4.5-4.7 TB/s (~40 VALU ops per 16 W
// bytes). This path moves the multiply onto v_mfma_scale_f32_16x16x128_f8f6f4
// (identity E8M0 scales; per-row float scales in the epilogue — the tile
// GEMM's numerics): per 2 KiB of weights a lane issues 2 NT loads + 2 x
// loads + 1 MFMA instead of ~250 VALU ops.
//
// Occupancy is the real constraint (a first cut with 32-row waves and no
// K-split ran 1.1 TB/s: rows/32 waves x 8 outstanding loads = 40 KiB in
// flight vs the ~3.6 MB the HBM latency x bandwidth product demands), so:
//   * 16-row groups (16x16x128: B has 16 batch columns >= M)
//   * split-K on grid.y sized so blocks ~ 8192 (KS in {1..16} dividing K/128)
//   * partial f32 strips in a device-resident workspace; the LAST block of
//     each row-group (atomic counter, released/acquired with threadfences)
//     reduces the strips IN ORDER (deterministic fp sum) and applies
//     wscale[row]*xscale[m], the residual add, or the silu(g)*u pairing.
// x is pre-quantized to fp8 by quant_fp8_kernel (or its fused-rmsnorm
// variant below) in the same stream — two launches per GEMV total.
typedef __attribute__((ext_vector_type(4))) float f32x4_v;

// rmsnorm + per-row fp8 quant (the NORM prologue of the VALU gemv, moved
// into the quant step so the stream kernel sees finished fp8 activations)
__global__ __launch_bounds__(256) void quant_norm_fp8_kernel(
    const uint32_t* __restrict__ x,   // [T, K/2] bf16x2
    const uint32_t* __restrict__ wn,  // [K/2] rmsnorm weight
    uint32_t* __restrict__ q,         // [T, K/4]
    float* __restrict__ scales,       // [T]
    int T, int K, float eps) {
    const int row = blockIdx.x * 4 + threadIdx.x / WAVE;
    if (row >= T) return;
    const int lane = threadIdx.x & (WAVE - 1);
    const int k2 = K / 2;
    const uint32_t* xr = x + (size_t)row * k2;
    float ss = 0.0f, amax = 1e-8f;
    for (int i = lane * 4; i < k2; i += WAVE * 4) {
        uint4 w = *reinterpret_cast<const uint4*>(xr + i);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            const float lo = bf16_lo((&w.x)[j]), hi = bf16_hi((&w.x)[j]);
            ss = fmaf(lo, lo, ss);
            ss = fmaf(hi, hi, ss);
            const uint32_t g = wn[i + j];
            amax = fmaxf(amax, fabsf(lo * bf16_lo(g)));
            amax = fmaxf(amax, fabsf(hi * bf16_hi(g)));
        }
    }
    ss = wave_reduce_sum(ss);
    amax = wave_reduce_max(amax);
    const float rstd = rsqrtf(ss / (float)K + eps);
    amax *= rstd;
    const float inv = FP8_MAX / amax;
    if (lane == 0) scales[row] = amax / FP8_MAX;
    uint32_t* qr = q + (size_t)row * (K / 4);
    for (int i = lane * 4; i < k2; i += WAVE * 4) {
        uint4 w = *reinterpret_cast<const uint4*>(xr + i);
        uint2 o;
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            const uint32_t g0 = wn[i + h * 2], g1 = wn[i + h * 2 + 1];
            unsigned packed = 0;
            packed = f32x2_to_fp8<false>(
                bf16_lo((&w.x)[h * 2]) * rstd * bf16_lo(g0) * inv,
                bf16_hi((&w.x)[h * 2]) * rstd * bf16_hi(g0) * inv, packed);
            packed = f32x2_to_fp8<true>(
                bf16_lo((&w.x)[h * 2 + 1]) * rstd * bf16_lo(g1) * inv,
                bf16_hi((&w.x)[h * 2 + 1]) * rstd * bf16_hi(g1) * inv, packed);
            (&o.x)[h] = packed;
        }
        *reinterpret_cast<uint2*>(qr + i / 2) = o;
    }
}

// Stream kernel: 256 threads; the FOUR waves split K (intra-block split-K:
// the f32 partial tiles meet in LDS — no cross-XCD fences, which made a
// grid-level split-K fixup 20x slower than the VALU path on this chip's
// per-XCD L2s). Wave w streams 16 W rows over k in [w*K/4, (w+1)*K/4) with
// a PF-deep register prefetch ring (PF=8 plain / 4 gateup: the in-flight
// byte product is what covers HBM latency — 16-row blocks x 4 waves x
// 256 B/lane ~ 40 MB chip-wide), x8 staged once to LDS.
template <bool GATEUP, bool ADDRES>
__global__ __launch_bounds__(256, 1) void gemv_fp8_mfma3_kernel(
    const uint32_t* __restrict__ x8,  // [M, K/4] fp8 (pre-quantized)
    const float* __restrict__ xsc,    // [M]
    const uint32_t* __restrict__ w8,  // [N, K/4] (gateup: [2I, K/4])
    const float* __restrict__ wscale, // [N] (gateup: [2I])
    uint32_t* __restrict__ out,       // [M, N] bf16 (gateup: [M, I])
    const uint32_t* __restrict__ res, // [M, N] residual (ADDRES)
    int M, int N, int K) {
    constexpr int PF = GATEUP ? 4 : 8;
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int fr = lane & 15;   // A row / B,D col
    const int fs = lane >> 4;   // 32-byte k-slice; D rows fs*4..+3
    const int rows = GATEUP ? N / 2 : N;
    const int I = rows;
    const int row0 = blockIdx.x * 16;
    const int wrow = min(row0 + fr, rows - 1);

    // x8 image -> LDS (whole [M, K] fp8), plus the 4-wave reduction tiles
    extern __shared__ __attribute__((aligned(16))) char xls[];
    const int kw = K / 4;  // words per row
    for (int i = tid * 4; i < M * kw; i += 256 * 4)
        *reinterpret_cast<uint4*>(xls + i * 4) =
            *reinterpret_cast<const uint4*>(x8 + i);
    __syncthreads();

    const int klen = K / 4;              // bytes per wave
    const int k0 = wid * klen;
    const int nk = klen / 128;           // 128-B MFMA steps
    const char* wp = reinterpret_cast<const char*>(w8);
    const uint32_t* wr0 = reinterpret_cast<const uint32_t*>(
        wp + (size_t)wrow * K + k0) + fs * 8;
    const uint32_t* wr1 = reinterpret_cast<const uint32_t*>(
        wp + (size_t)(wrow + (GATEUP ? I : 0)) * K + k0) + fs * 8;
    const int xrow = fr < M ? fr : 0;
    const char* xr = xls + (size_t)xrow * K + k0 + fs * 32;

    f32x4_v acc0 = {}, acc1 = {};
    i32x8_f8 abuf[PF], ubuf[GATEUP ? PF : 1];
    auto load_a = [&](int p, int kb) {
        *reinterpret_cast<u32x4*>(&abuf[p]) = nt_load4f(wr0 + kb * 32);
        *(reinterpret_cast<u32x4*>(&abuf[p]) + 1) =
            nt_load4f(wr0 + kb * 32 + 4);
        if (GATEUP) {
            *reinterpret_cast<u32x4*>(&ubuf[p]) = nt_load4f(wr1 + kb * 32);
            *(reinterpret_cast<u32x4*>(&ubuf[p]) + 1) =
                nt_load4f(wr1 + kb * 32 + 4);
        }
    };
#pragma unroll
    for (int p = 0; p < PF; ++p) load_a(p, p);
    // x fragment double-buffered ONE step ahead (an inline ds_read feeding
    // its own MFMA serializes on lgkmcnt(0) every step)
    i32x8_f8 bcur;
    *reinterpret_cast<uint4*>(&bcur) = *reinterpret_cast<const uint4*>(xr);
    *(reinterpret_cast<uint4*>(&bcur) + 1) =
        *reinterpret_cast<const uint4*>(xr + 16);
    for (int kb0 = 0; kb0 < nk; kb0 += PF) {
        const bool more = kb0 + PF < nk;
#pragma unroll
        for (int p = 0; p < PF; ++p) {
            const int kb = kb0 + p;
            i32x8_f8 bnext;
            if (kb + 1 < nk) {
                *reinterpret_cast<uint4*>(&bnext) =
                    *reinterpret_cast<const uint4*>(xr + (kb + 1) * 128);
                *(reinterpret_cast<uint4*>(&bnext) + 1) =
                    *reinterpret_cast<const uint4*>(xr + (kb + 1) * 128 + 16);
            }
            acc0 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
                abuf[p], bcur, acc0, 0, 0, 0, 0x7f, 0, 0x7f);
            if (GATEUP)
                acc1 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
                    ubuf[p], bcur, acc1, 0, 0, 0, 0x7f, 0, 0x7f);
            if (more) load_a(p, kb0 + PF + p);
            if (kb + 1 < nk) bcur = bnext;
        }
    }

    // intra-block reduction: [wave][plane][col fr][row fs*4+i] f32 in LDS
    __syncthreads();  // image reads done before the tiles overwrite it
    float* red = reinterpret_cast<float*>(xls);
    const int planes = GATEUP ? 2 : 1;
    {
        float* t = red + ((wid * planes) * 16 + fr) * 16 + fs * 4;
        *reinterpret_cast<f32x4_v*>(t) = acc0;
        if (GATEUP)
            *reinterpret_cast<f32x4_v*>(t + 16 * 16) = acc1;
    }
    __syncthreads();
    if (wid != 0 || fr >= M) return;
    const int m = fr;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        const int row = row0 + fs * 4 + i;
        if (row >= rows) continue;
        float sg = 0.0f, su = 0.0f;
#pragma unroll
        for (int w = 0; w < 4; ++w) {
            sg += red[((w * planes) * 16 + m) * 16 + fs * 4 + i];
            if (GATEUP)
                su += red[((w * planes + 1) * 16 + m) * 16 + fs * 4 + i];
        }
        float v;
        if (GATEUP) {
            const float gg = sg * wscale[row] * xsc[m];
            const float uu = su * wscale[row + I] * xsc[m];
            v = gg / (1.0f + __expf(-gg)) * uu;
        } else {
            v = sg * wscale[row] * xsc[m];
            if (ADDRES)
                v += bf16_to_f32(reinterpret_cast<const uint16_t*>(
                    res)[(size_t)m * rows + row]);
        }
        reinterpret_cast<uint16_t*>(out)[(size_t)m * rows + row] =
            f32_to_bf16(v);
    }
}

// device scratch for the pre-quantized x. Allocated ONCE on first use —
// callers must make a first (eager) call before any hipGraph capture; the
// engine's weight-quantization path does (ops.quant_fp8 wrapper).
static uint32_t* g_gemv8_x8 = nullptr;
static float* g_gemv8_xsc = nullptr;
#define GEMV8_MAX_M 16
#define GEMV8_MAX_K 32768

extern "C" int oa_fp8_gemv_scratch_init(void) {
    if (g_gemv8_x8) return 0;
    if (hipMalloc(&g_gemv8_x8, (size_t)GEMV8_MAX_M * GEMV8_MAX_K) != hipSuccess)
        return -1;
    if (hipMalloc(&g_gemv8_xsc, GEMV8_MAX_M * 4) != hipSuccess) return -1;
    return 0;
}

extern "C" int oa_gemv_fp8_mfma(void* stream, const void* x, const void* w8,
                                const void* wscale, void* out, const void* wn,
                                const void* res, int M, int N, int K,
                                float eps, int mode, int gateup) {
    const int rows = gateup ? N / 2 : N;
    if (K % 512 != 0 || (K / 512) % (gateup ? 4 : 8) != 0 ||
        M > GEMV8_MAX_M || K > GEMV8_MAX_K)
        return -100;
    const size_t img = (size_t)M * K;
    if (img > 128 * 1024) return -101;  // x image must fit LDS
    if (!g_gemv8_x8 && oa_fp8_gemv_scratch_init() != 0) return -102;
    hipStream_t s = (hipStream_t)stream;
    if (mode & 1) {
        hipLaunchKernelGGL(quant_norm_fp8_kernel, dim3(CEIL_DIV(M, 4)),
                           dim3(256), 0, s, (const uint32_t*)x,
                           (const uint32_t*)wn, g_gemv8_x8, g_gemv8_xsc, M, K,
                           eps);
    } else {
        hipLaunchKernelGGL(quant_fp8_kernel, dim3(CEIL_DIV(M, 4)), dim3(256),
                           0, s, (const uint32_t*)x, g_gemv8_x8, g_gemv8_xsc,
                           M, K);
    }
    const size_t shmem =
        img > (size_t)(gateup ? 2 : 1) * 4 * 16 * 16 * 4 * 4
            ? img
            : (size_t)(gateup ? 2 : 1) * 4 * 16 * 16 * 4 * 4;
    dim3 grid(CEIL_DIV(rows, 16)), block(256);
#define LAUNCH_MF3(GUV, RESV)                                                  \
    hipLaunchKernelGGL((gemv_fp8_mfma3_kernel<GUV, RESV>), grid, block, shmem, \
                       s, g_gemv8_x8, g_gemv8_xsc, (const uint32_t*)w8,        \
                       (const float*)wscale, (uint32_t*)out,                   \
                       (const uint32_t*)res, M, N, K)
    if (gateup) {
        LAUNCH_MF3(true, false);
    } else if (mode & 2) {
        LAUNCH_MF3(false, true);
    } else {
        LAUNCH_MF3(false, false);
    }
#undef LAUNCH_MF3
    HIP_CHECK_LAUNCH();
    return 0;
}
