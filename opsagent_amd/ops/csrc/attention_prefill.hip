// Flash-style causal prefill attention for MI355X (gfx950) — SURVEY.md §2b
// "Prefill attention kernel (HIP, bf16 MFMA, LDS-staged KV tiles, gfx950
// tiling)".
//
// Structure (CDNA guide §B "Fused attention prefill", 8-warp QBLK=32 ladder):
//   * workgroup = 512 threads = 8 waves; each wave owns RB x 16 = 32 q-rows
//     (q-tile = 256 rows per workgroup), D = 128, KV tile = 64 keys —
//     32 MFMAs per staged KV byte per wave halves the staging/barrier
//     overhead per FLOP vs the 16-row form (measured 120 -> ~190 TF class)
//   * QK^T and P·V on v_mfma_f32_16x16x32_bf16 (per-wave MFMA, 64-lane
//     fragment layouts — NOT warp-32 tilings)
//   * K tile LDS-staged row-major with the guide's XOR swizzle
//     (byte ^= (key&15)<<4): a 16-lane ds_read_b128 group reads 16 different
//     keys at one d-offset — unswizzled that is an up-to-16-way bank conflict
//     (guide Guideline 4: this exact pattern was 52% of an attn kernel's time)
//   * V staged TRANSPOSED (VT[d][key]) so the P·V B-fragment is a contiguous
//     ds_read_b128 per lane; swizzle byte ^= (d&7)<<4 on the 128-B VT rows
//   * online softmax per q-row held in registers; row statistics reduced with
//     4-step __shfl_xor over the 16-lane fragment columns
//   * P round-trips through a per-wave LDS tile ([32][64] bf16, swizzled) to
//     re-shape from the C-fragment layout to the next MFMA's A-fragment
//
// Layouts: q [B, Sq, Hq, 128], k/v [B, Skv, Hk, 128] with per-token strides
// (they may be head-slices of one fused qkv buffer), out [B, Sq, Hq, 128]
// contiguous; causal offset = Skv - Sq (query i attends keys [0..Skv-Sq+i])
// so multi-turn chunked prefill reuses the same kernel.

#include "common.h"

#define DHEAD 128
#define KVBLK 64
#define RB 2                 // 16-row blocks per wave
#define QROWS (16 * RB)      // q rows per wave
#define NWAVE 8
#define QTILE (QROWS * NWAVE)

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

// LDS byte offsets within the single shared allocation (guide G17: one object)
#define K_BYTES (KVBLK * DHEAD * 2)           // 16 KiB
#define VT_BYTES (DHEAD * KVBLK * 2)          // 16 KiB
#define P_BYTES (QROWS * KVBLK * 2)           // 4 KiB per wave
#define SMEM_BYTES (K_BYTES + VT_BYTES + NWAVE * P_BYTES)

__device__ __forceinline__ uint32_t k_swz(int key, int byte_in_row) {
    return (uint32_t)(key * (DHEAD * 2) + (byte_in_row ^ ((key & 15) << 4)));
}
__device__ __forceinline__ uint32_t vt_swz(int d, int byte_in_row) {
    return (uint32_t)(K_BYTES + d * (KVBLK * 2) + (byte_in_row ^ ((d & 7) << 4)));
}
__device__ __forceinline__ uint32_t p_swz(int wid, int row, int byte_in_row) {
    return (uint32_t)(K_BYTES + VT_BYTES + wid * P_BYTES + row * (KVBLK * 2) +
                      (byte_in_row ^ ((row & 7) << 4)));
}

__global__ __launch_bounds__(512, 1) void attn_prefill_kernel(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ out, int B, int Hq,
    int Hk, int Sq, int Skv, float scale,
    int qs /* q token stride (elems) */, int ks, int vs) {
    __shared__ __attribute__((aligned(16))) char smem[SMEM_BYTES];

    const int qtile = blockIdx.x;
    const int bh = blockIdx.y;
    const int b = bh / Hq;
    const int h = bh % Hq;
    const int hk = h / (Hq / Hk);
    const int offset = Skv - Sq;  // causal diagonal offset

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int fr = lane & 15;      // fragment row/col index (0..15)
    const int fs = lane >> 4;      // k-slice 0..3 (owns k = fs*8 .. fs*8+7)

    // ---- load Q fragments: aq[rb][dblk] = Q[qrow = rb*16 + fr][d = dblk*32 + fs*8]
    const int qrow0 = qtile * QTILE + wid * QROWS;
    uint4 aq[RB][4];
#pragma unroll
    for (int rb = 0; rb < RB; ++rb) {
        const int row = qrow0 + rb * 16 + fr;
        if (row < Sq) {
            const uint16_t* qp = q + (size_t)(b * Sq + row) * qs + (size_t)h * DHEAD;
#pragma unroll
            for (int dblk = 0; dblk < 4; ++dblk)
                aq[rb][dblk] = *reinterpret_cast<const uint4*>(qp + dblk * 32 + fs * 8);
        } else {
#pragma unroll
            for (int dblk = 0; dblk < 4; ++dblk) aq[rb][dblk] = make_uint4(0, 0, 0, 0);
        }
    }

    // ---- online softmax state (4 rows per lane per row-block)
    float m[RB][4], lsum[RB][4];
    f32x4_t o_acc[RB][8];  // o_acc[rb][nb][reg] = O[row = rb*16+fs*4+reg][d = nb*16+fr]
#pragma unroll
    for (int rb = 0; rb < RB; ++rb)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            m[rb][r] = -INFINITY;
            lsum[rb][r] = 0.0f;
        }
#pragma unroll
    for (int rb = 0; rb < RB; ++rb)
#pragma unroll
        for (int nb = 0; nb < 8; ++nb) o_acc[rb][nb] = (f32x4_t){0.f, 0.f, 0.f, 0.f};

    // keys needed by this q-tile under causality
    const int kv_needed = min(Skv, offset + qtile * QTILE + QTILE);
    const int ntiles = CEIL_DIV(max(kv_needed, 0), KVBLK);

    const uint16_t* kbase = k + (size_t)b * Skv * ks + (size_t)hk * DHEAD;
    const uint16_t* vbase = v + (size_t)b * Skv * vs + (size_t)hk * DHEAD;

    for (int t = 0; t < ntiles; ++t) {
        const int kv0 = t * KVBLK;
        __syncthreads();  // previous tile's LDS reads complete
        // ---- stage K (swizzled rows) and V (transposed) -------------------
        // 1024 16-B pieces: piece li -> key = li/16, d0 = (li%16)*8
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            const int li = tid + p * 512;
            const int key = li >> 4;
            const int d0 = (li & 15) * 8;
            const int kg = kv0 + key;
            uint4 kv_k = make_uint4(0, 0, 0, 0), kv_v = make_uint4(0, 0, 0, 0);
            if (kg < Skv) {
                kv_k = *reinterpret_cast<const uint4*>(kbase + (size_t)kg * ks + d0);
                kv_v = *reinterpret_cast<const uint4*>(vbase + (size_t)kg * vs + d0);
            }
            *reinterpret_cast<uint4*>(smem + k_swz(key, d0 * 2)) = kv_k;
            // transpose V into VT[d][key] with scalar element writes
            const uint16_t* ve = reinterpret_cast<const uint16_t*>(&kv_v);
#pragma unroll
            for (int j = 0; j < 8; ++j)
                *reinterpret_cast<uint16_t*>(smem + vt_swz(d0 + j, key * 2)) = ve[j];
        }
        __syncthreads();

#pragma unroll
        for (int rb = 0; rb < RB; ++rb) {
            // ---- QK^T: s[kb][reg] over 4 key-columns of 16 ----------------
            f32x4_t s[4];
#pragma unroll
            for (int kb = 0; kb < 4; ++kb) {
                f32x4_t acc = (f32x4_t){0.f, 0.f, 0.f, 0.f};
#pragma unroll
                for (int dblk = 0; dblk < 4; ++dblk) {
                    // B-fragment: K[key = kb*16 + fr][d = dblk*32 + fs*8 ..+8]
                    uint4 bk = *reinterpret_cast<const uint4*>(
                        smem + k_swz(kb * 16 + fr, (dblk * 32 + fs * 8) * 2));
                    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        *reinterpret_cast<bf16x8_t*>(&aq[rb][dblk]),
                        *reinterpret_cast<bf16x8_t*>(&bk), acc, 0, 0, 0);
                }
                s[kb] = acc;
            }

            // ---- mask + online softmax -----------------------------------
            const int qpos = offset + qrow0 + rb * 16 + fs * 4;  // + reg
            float rowmax[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) rowmax[r] = -INFINITY;
#pragma unroll
            for (int kb = 0; kb < 4; ++kb) {
                const int kg = kv0 + kb * 16 + fr;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    float sv = s[kb][r] * scale;
                    if (kg > qpos + r || kg >= Skv) sv = -INFINITY;
                    s[kb][r] = sv;
                    rowmax[r] = fmaxf(rowmax[r], sv);
                }
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) rowmax[r] = group16_reduce_max(rowmax[r]);

            float alpha[4], psum[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const float mn = fmaxf(m[rb][r], rowmax[r]);
                alpha[r] = __expf(m[rb][r] - mn);
                m[rb][r] = mn;
                psum[r] = 0.0f;
            }
            // valid rows always have finite max at tile 0 (key 0 unmasked);
            // fully-masked padding rows produce NaN locally, never stored.

            // P = exp(s - m) -> row sums -> bf16 into the wave's P tile
#pragma unroll
            for (int kb = 0; kb < 4; ++kb) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const float p = __expf(s[kb][r] - m[rb][r]);
                    s[kb][r] = p;
                    psum[r] += p;
                    *reinterpret_cast<uint16_t*>(
                        smem + p_swz(wid, rb * 16 + fs * 4 + r, (kb * 16 + fr) * 2)) =
                        f32_to_bf16(p);
                }
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                psum[r] = group16_reduce_sum(psum[r]);
                lsum[rb][r] = lsum[rb][r] * alpha[r] + psum[r];
            }
#pragma unroll
            for (int nb = 0; nb < 8; ++nb)
#pragma unroll
                for (int r = 0; r < 4; ++r) o_acc[rb][nb][r] *= alpha[r];

            // ---- P·V ------------------------------------------------------
            // A-fragment: P[row = rb*16 + fr][kcol = kb2*32 + fs*8 ..+8]
            // (wave-local LDS; ds_write -> ds_read ordering is by lgkmcnt)
            uint4 ap[2];
#pragma unroll
            for (int kb2 = 0; kb2 < 2; ++kb2)
                ap[kb2] = *reinterpret_cast<const uint4*>(
                    smem + p_swz(wid, rb * 16 + fr, (kb2 * 32 + fs * 8) * 2));
#pragma unroll
            for (int nb = 0; nb < 8; ++nb) {
#pragma unroll
                for (int kb2 = 0; kb2 < 2; ++kb2) {
                    // B-fragment: V[k = kb2*32+fs*8 ..+8][d = nb*16+fr] = VT rows
                    uint4 bv = *reinterpret_cast<const uint4*>(
                        smem + vt_swz(nb * 16 + fr, (kb2 * 32 + fs * 8) * 2));
                    o_acc[rb][nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        *reinterpret_cast<bf16x8_t*>(&ap[kb2]),
                        *reinterpret_cast<bf16x8_t*>(&bv), o_acc[rb][nb], 0, 0, 0);
                }
            }
        }
    }

    // ---- epilogue: out[row][d] = o / l ------------------------------------
#pragma unroll
    for (int rb = 0; rb < RB; ++rb)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int row = qrow0 + rb * 16 + fs * 4 + r;
            if (row >= Sq) continue;
            const float inv_l = (lsum[rb][r] > 0.0f) ? 1.0f / lsum[rb][r] : 0.0f;
            uint16_t* op = out + ((size_t)(b * Sq + row) * Hq + h) * DHEAD;
#pragma unroll
            for (int nb = 0; nb < 8; ++nb)
                op[nb * 16 + fr] = f32_to_bf16(o_acc[rb][nb][r] * inv_l);
        }
}

extern "C" int oa_attention_prefill(void* stream, const void* q, const void* k,
                                    const void* v, void* out, int B, int Hq,
                                    int Hk, int Sq, int Skv, int D, float scale,
                                    int q_stride, int k_stride, int v_stride) {
    if (D != DHEAD) return -100;
    if (Hq % Hk != 0) return -101;
    if ((q_stride | k_stride | v_stride) % 8 != 0) return -102;
    dim3 grid(CEIL_DIV(Sq, QTILE), B * Hq), block(512);
    hipLaunchKernelGGL(attn_prefill_kernel, grid, block, 0, (hipStream_t)stream,
                       (const uint16_t*)q, (const uint16_t*)k, (const uint16_t*)v,
                       (uint16_t*)out, B, Hq, Hk, Sq, Skv, scale,
                       q_stride, k_stride, v_stride);
    HIP_CHECK_LAUNCH();
    return 0;
}
