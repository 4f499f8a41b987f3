// Paged decode attention for MI355X (gfx950) — SURVEY.md §2b "Decode (paged
// KV) attention kernel".
//
// Single new token per sequence attends to the whole paged KV cache. The op
// is HBM-bound (reads n_tokens * 2 * Hk * D * 2 bytes); the design follows
// the CDNA guide Appendix B "Attention decode": vectorized bf16 KV loads
// (16 B/lane), shuffle-reduce dot products, online softmax in registers, and
// sequence-split ("split-K over keys") so the grid covers the chip even at
// batch 1 (B*Hk workgroups alone would leave 256 CUs mostly idle).
//
// Layouts:
//   q        [B, Hq, 128] bf16
//   kc/vc    flat [num_slots, Hk, 128] bf16 (slot = block_id*block_size + off)
//   block_table [B, max_blocks] int32
//   seq_lens [B] int32 (tokens in cache, incl. the current token)
//   o_part   [B*Hk*NSPLIT, G, 128] f32, ml_part [B*Hk*NSPLIT, G, 2] f32
//   out      [B, Hq, 128] bf16
//
// Wave layout: 16 lanes per key row (16 x 8 dims = 128), 4 keys in flight per
// wave, 4 waves per workgroup each owning a quarter of the split's key range.

#include "common.h"

#define DHEAD 128

// FUSED = true folds the split-combine into this launch with the CDNA guide's
// Guideline-16 in-launch split-K hand-off (plain-store form): each split
// block publishes its partial with an agent-scope release and bumps a ticket
// counter; the LAST arriver for (b, h) acquires, reads every split's slab
// and writes the final output — saving the second kernel launch per layer.
// `cnt` must be zeroed before every launch by a hipMemsetAsync on the stream
// (a memset node under graph capture, replayed first — guide G16).
// ROPE = true additionally fuses the per-token RoPE + paged-KV scatter that
// the separate rope_kv kernel did (4.7 us/layer of pure launch latency at
// decode): every block ropes its G q heads IN REGISTERS (rotate-half partner
// dims live in lane dl^8 — one __shfl_xor), the main loop covers only the
// CACHED keys [0, n-1), and the split that owns key n-1 ropes the new k from
// the raw GEMV output, scatters roped-k and v to the cache slot, and folds
// the new key into its online softmax. kin/vin/cs/sn/slots/kcw/vcw are only
// read when ROPE.
template <int G, bool FUSED, bool ROPE, bool DEEP = true>
__global__ __launch_bounds__(256) void decode_attn_kernel(
    const uint32_t* __restrict__ q, const uint32_t* __restrict__ kc,
    const uint32_t* __restrict__ vc, const int* __restrict__ block_table,
    const int* __restrict__ seq_lens, float* __restrict__ o_part,
    float* __restrict__ ml_part, unsigned* __restrict__ cnt,
    uint32_t* __restrict__ out, int B, int Hk, int max_blocks,
    int block_shift /* log2(block_size) */, int nsplit, float scale,
    int qs2 /* q batch-row stride in words */,
    const uint32_t* __restrict__ kin, const uint32_t* __restrict__ vin,
    const float* __restrict__ cs, const float* __restrict__ sn,
    const int* __restrict__ slots, uint32_t* __restrict__ kcw,
    uint32_t* __restrict__ vcw, int ks2, int vs2) {
    const int bh = blockIdx.x;
    const int b = bh / Hk;
    const int h = bh % Hk;
    const int split = blockIdx.y;
    const int Hq = Hk * G;

    const int n = seq_lens[b];
    const int chunk = CEIL_DIV(n, nsplit);
    const int kstart = split * chunk;
    const int kend = min(n, kstart + chunk);
    // ROPE: the new key (index n-1) is handled from registers, not cache
    const int n_cached = ROPE ? (n - 1) : n;
    const int kend_c = min(kend, n_cached);

    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int g16 = lane >> 4;   // key slot within the wave's 4-key window
    const int dl = lane & 15;    // owns dims dl*8 .. dl*8+7

    const int part_idx = (bh * nsplit + split);
    float* opart_row = o_part + ((size_t)part_idx * G) * DHEAD;
    float* mlpart_row = ml_part + ((size_t)part_idx * G) * 2;

    // per-wave sub-range (over CACHED keys only)
    const int span = kend_c - kstart;
    const int wchunk = CEIL_DIV(max(span, 0), 4);
    const int wstart = kstart + wid * wchunk;
    const int wend = min(kend_c, wstart + wchunk);

    // load q fragments: [G][8] floats per lane (all lane groups redundant)
    float qreg[G][8];
#pragma unroll
    for (int gh = 0; gh < G; ++gh) {
        const uint32_t* qp = q + (size_t)b * qs2 + ((size_t)(h * G + gh) * DHEAD + dl * 8) / 2;
        uint4 w = *reinterpret_cast<const uint4*>(qp);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            qreg[gh][j * 2] = bf16_lo((&w.x)[j]);
            qreg[gh][j * 2 + 1] = bf16_hi((&w.x)[j]);
        }
    }

    // ROPE: rotate-half q in registers. Lane dl owns dims dl*8..dl*8+7; the
    // partner half (d +/- 64) lives in lane dl^8 of the same 16-lane group.
    float c8[8], s8[8];
    if (ROPE) {
        const int pos = n - 1;
        const float* crow = cs + (size_t)pos * (DHEAD / 2) + (dl & 7) * 8;
        const float* srow = sn + (size_t)pos * (DHEAD / 2) + (dl & 7) * 8;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            c8[j] = crow[j];
            s8[j] = srow[j];
        }
#pragma unroll
        for (int gh = 0; gh < G; ++gh) {
            float part[8];
#pragma unroll
            for (int j = 0; j < 8; ++j)
                part[j] = __shfl_xor(qreg[gh][j], 8, WAVE);
#pragma unroll
            for (int j = 0; j < 8; ++j)
                qreg[gh][j] = (dl < 8)
                                  ? qreg[gh][j] * c8[j] - part[j] * s8[j]
                                  : qreg[gh][j] * c8[j] + part[j] * s8[j];
        }
    }

    float m[G], lsum[G], o[G][8];
#pragma unroll
    for (int gh = 0; gh < G; ++gh) {
        m[gh] = -INFINITY;
        lsum[gh] = 0.0f;
#pragma unroll
        for (int j = 0; j < 8; ++j) o[gh][j] = 0.0f;
    }

    const int bmask = (1 << block_shift) - 1;
    const int* btrow = block_table + (size_t)b * max_blocks;

    // 2-deep key-quad pipeline: issue the raw 16-B K/V loads for quad i+1
    // before the softmax/accumulate work of quad i — the online-softmax
    // update is loop-carried but the loads are not, and one quad in flight
    // (2 x 16 B/lane) is far too little to cover HBM latency at 8 waves/CU.
    auto issue_loads = [&](int kk0, uint4& kw, uint4& vw, bool& valid, int64_t&) {
        const int kk = kk0 + g16;
        valid = (kk0 < wend) && (kk < wend);
        if (valid) {
            const int64_t slot =
                (int64_t)btrow[kk >> block_shift] << block_shift | (kk & bmask);
            kw = *reinterpret_cast<const uint4*>(
                kc + ((size_t)slot * Hk + h) * (DHEAD / 2) + dl * 4);
            vw = *reinterpret_cast<const uint4*>(
                vc + ((size_t)slot * Hk + h) * (DHEAD / 2) + dl * 4);
        }
    };
    auto process = [&](const uint4& kw, const uint4& vw, bool valid) {
        float kv_k[8], kv_v[8];
        if (valid) {
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                kv_k[j * 2] = bf16_lo((&kw.x)[j]);
                kv_k[j * 2 + 1] = bf16_hi((&kw.x)[j]);
                kv_v[j * 2] = bf16_lo((&vw.x)[j]);
                kv_v[j * 2 + 1] = bf16_hi((&vw.x)[j]);
            }
        }
#pragma unroll
        for (int gh = 0; gh < G; ++gh) {
            float s = 0.0f;
            if (valid) {
#pragma unroll
                for (int j = 0; j < 8; ++j) s += qreg[gh][j] * kv_k[j];
            }
            s = group16_reduce_sum(s);  // all 16 lanes of the group get the dot
            if (valid) {
                s *= scale;
                const float mn = fmaxf(m[gh], s);
                const float alpha = __expf(m[gh] - mn);
                const float p = __expf(s - mn);
#pragma unroll
                for (int j = 0; j < 8; ++j) o[gh][j] = o[gh][j] * alpha + p * kv_v[j];
                lsum[gh] = lsum[gh] * alpha + p;
                m[gh] = mn;
            }
        }
    };
    {
        // 4-deep key-quad pipeline (modulo-scheduled with NAMED slots —
        // a runtime-indexed ring would go to scratch, guide rule 20):
        // 2-deep left most of the ~900-cycle HBM latency exposed at small
        // batch, where one quad is only 32 B/lane in flight. DEEP is
        // dispatched for B<=4 only: the extra slots cost ~34 VGPRs
        // (G4: 140->174 -> one fewer wave/SIMD) — right when latency-
        // bound, an occupancy tax once a big batch saturates HBM.
        // G8 keeps depth 2 (qreg+o already hold 128 VGPRs).
        int64_t dummy = 0;
        if (G <= 4 && DEEP) {
            uint4 kA{}, vA4{}, kB{}, vB4{}, kC{}, vC4{}, kD{}, vD4{};
            bool va = false, vb = false, vc2 = false, vd = false;
            issue_loads(wstart, kA, vA4, va, dummy);
            issue_loads(wstart + 4, kB, vB4, vb, dummy);
            issue_loads(wstart + 8, kC, vC4, vc2, dummy);
            for (int kk0 = wstart; kk0 < wend; kk0 += 16) {
                issue_loads(kk0 + 12, kD, vD4, vd, dummy);
                process(kA, vA4, va);
                issue_loads(kk0 + 16, kA, vA4, va, dummy);
                process(kB, vB4, vb);
                issue_loads(kk0 + 20, kB, vB4, vb, dummy);
                process(kC, vC4, vc2);
                issue_loads(kk0 + 24, kC, vC4, vc2, dummy);
                process(kD, vD4, vd);
            }
        } else {
            uint4 kwA{}, vwA{}, kwB{}, vwB{};
            bool va = false, vb = false;
            issue_loads(wstart, kwA, vwA, va, dummy);
            for (int kk0 = wstart; kk0 < wend; kk0 += 4) {
                issue_loads(kk0 + 4, kwB, vwB, vb, dummy);
                process(kwA, vwA, va);
                kwA = kwB; vwA = vwB; va = vb;
            }
        }
    }

    // ROPE: the owning split ropes the NEW key from the raw GEMV output,
    // scatters roped-k and v to the cache slot, and folds it into wave 3's
    // online softmax (the cross-wave LDS merge below absorbs it).
    if (ROPE) {
        const int pos = n - 1;
        if (pos >= kstart && pos < kend && wid == 3) {
            const bool nv = (g16 == 0);
            uint4 kw{}, vw{};
            if (nv) {
                kw = *reinterpret_cast<const uint4*>(
                    kin + (size_t)b * ks2 + ((size_t)h * DHEAD + dl * 8) / 2);
                vw = *reinterpret_cast<const uint4*>(
                    vin + (size_t)b * vs2 + ((size_t)h * DHEAD + dl * 8) / 2);
            }
            float kf[8], vf[8];
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                kf[j * 2] = bf16_lo((&kw.x)[j]);
                kf[j * 2 + 1] = bf16_hi((&kw.x)[j]);
                vf[j * 2] = bf16_lo((&vw.x)[j]);
                vf[j * 2 + 1] = bf16_hi((&vw.x)[j]);
            }
            float kp[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) kp[j] = __shfl_xor(kf[j], 8, WAVE);
            float kr[8];
#pragma unroll
            for (int j = 0; j < 8; ++j)
                kr[j] = (dl < 8) ? kf[j] * c8[j] - kp[j] * s8[j]
                                 : kf[j] * c8[j] + kp[j] * s8[j];
            if (nv) {
                const int64_t slot = slots[b];
                uint4 kout;
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    (&kout.x)[j] = pack_bf16x2(kr[j * 2], kr[j * 2 + 1]);
                *reinterpret_cast<uint4*>(
                    kcw + ((size_t)slot * Hk + h) * (DHEAD / 2) + dl * 4) = kout;
                *reinterpret_cast<uint4*>(
                    vcw + ((size_t)slot * Hk + h) * (DHEAD / 2) + dl * 4) = vw;
            }
#pragma unroll
            for (int gh = 0; gh < G; ++gh) {
                float s = 0.0f;
                if (nv) {
#pragma unroll
                    for (int j = 0; j < 8; ++j) s += qreg[gh][j] * kr[j];
                }
                s = group16_reduce_sum(s);
                if (nv) {
                    s *= scale;
                    const float mn = fmaxf(m[gh], s);
                    const float alpha = __expf(m[gh] - mn);
                    const float p = __expf(s - mn);
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        o[gh][j] = o[gh][j] * alpha + p * vf[j];
                    lsum[gh] = lsum[gh] * alpha + p;
                    m[gh] = mn;
                }
            }
        }
    }

    // merge the wave's 4 key-groups (lanes l, l^16, l^32 hold same dims)
#pragma unroll
    for (int off = 16; off <= 32; off <<= 1) {
#pragma unroll
        for (int gh = 0; gh < G; ++gh) {
            const float m2 = __shfl_xor(m[gh], off, WAVE);
            const float l2 = __shfl_xor(lsum[gh], off, WAVE);
            float o2[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) o2[j] = __shfl_xor(o[gh][j], off, WAVE);
            const float mn = fmaxf(m[gh], m2);
            // guard: empty partial (l == 0, m == -inf) must not produce NaN
            const float a1 = (lsum[gh] > 0.0f) ? __expf(m[gh] - mn) : 0.0f;
            const float a2 = (l2 > 0.0f) ? __expf(m2 - mn) : 0.0f;
#pragma unroll
            for (int j = 0; j < 8; ++j) o[gh][j] = o[gh][j] * a1 + o2[j] * a2;
            lsum[gh] = lsum[gh] * a1 + l2 * a2;
            m[gh] = (lsum[gh] > 0.0f) ? mn : -INFINITY;
        }
    }

    // cross-wave combine through LDS
    __shared__ float lds_o[4][G][DHEAD];
    __shared__ float lds_ml[4][G][2];
    if (lane < 16) {
#pragma unroll
        for (int gh = 0; gh < G; ++gh) {
#pragma unroll
            for (int j = 0; j < 8; ++j) lds_o[wid][gh][dl * 8 + j] = o[gh][j];
            if (dl == 0) {
                lds_ml[wid][gh][0] = m[gh];
                lds_ml[wid][gh][1] = lsum[gh];
            }
        }
    }
    __syncthreads();

    // 256 threads cover G*128 outputs
    for (int idx = threadIdx.x; idx < G * DHEAD; idx += blockDim.x) {
        const int gh = idx / DHEAD;
        const int d = idx % DHEAD;
        float mt = -INFINITY;
#pragma unroll
        for (int w = 0; w < 4; ++w)
            if (lds_ml[w][gh][1] > 0.0f) mt = fmaxf(mt, lds_ml[w][gh][0]);
        float lt = 0.0f, ot = 0.0f;
#pragma unroll
        for (int w = 0; w < 4; ++w) {
            const float lw = lds_ml[w][gh][1];
            if (lw > 0.0f) {
                const float sc = __expf(lds_ml[w][gh][0] - mt);
                lt += lw * sc;
                ot += lds_o[w][gh][d] * sc;
            }
        }
        opart_row[gh * DHEAD + d] = ot;
        if (d == 0) {
            mlpart_row[gh * 2 + 0] = mt;
            mlpart_row[gh * 2 + 1] = lt;
        }
    }

    if (!FUSED) return;

    // ---- G16 publish + reducer election (plain-store recipe) --------------
    // every wave drains its slab stores, then one lane releases and takes a
    // ticket; the block that draws nsplit-1 becomes the reducer
    __shared__ int s_last;
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // EVERY wave
    __syncthreads();
    if (threadIdx.x == 0) {
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
        // restate the post-wbl2 wait the compiler may drop (guide pitfall 12)
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        const unsigned prev = __hip_atomic_fetch_add(
            &cnt[bh], 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        s_last = (prev == (unsigned)(nsplit - 1));
    }
    __syncthreads();
    if (!s_last) return;
    if (threadIdx.x == 0)
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");  // drop stale L1
    __syncthreads();

    // ---- reduce all splits of (b, h) and write the final O ----------------
    const int Hq2 = Hk * G;
    const int base = bh * nsplit;
    for (int idx = threadIdx.x; idx < G * DHEAD; idx += blockDim.x) {
        const int gh = idx / DHEAD;
        const int d = idx % DHEAD;
        float mt = -INFINITY;
        for (int s = 0; s < nsplit; ++s) {
            const float lw = ml_part[((size_t)(base + s) * G + gh) * 2 + 1];
            if (lw > 0.0f) mt = fmaxf(mt, ml_part[((size_t)(base + s) * G + gh) * 2]);
        }
        float lt = 0.0f, ot = 0.0f;
        for (int s = 0; s < nsplit; ++s) {
            const float mw = ml_part[((size_t)(base + s) * G + gh) * 2];
            const float lw = ml_part[((size_t)(base + s) * G + gh) * 2 + 1];
            if (lw > 0.0f) {
                const float sc = __expf(mw - mt);
                lt += lw * sc;
                ot += o_part[((size_t)(base + s) * G + gh) * DHEAD + d] * sc;
            }
        }
        const float res = (lt > 0.0f) ? ot / lt : 0.0f;
        reinterpret_cast<uint16_t*>(out)[(size_t)(b * Hq2 + h * G + gh) * DHEAD + d] =
            f32_to_bf16(res);
    }
}

// Combine split partials -> final output. Grid: B*Hq blocks, 512 threads:
// thread (d = tid & 127, sh = tid >> 7) accumulates splits sh, sh+4, ... so
// the split loop runs nsplit/4 iterations with 4-way ILP, then the 4 partial
// (m, l, o_d) triples merge through LDS. The (m, l) table is staged into LDS
// once by the first wave (the naive one-thread-one-d loop over 64 splits was
// 32 us/launch — 3x the attention kernel itself).
template <int G>
__global__ __launch_bounds__(512) void decode_combine_kernel(
    const float* __restrict__ o_part, const float* __restrict__ ml_part,
    uint32_t* __restrict__ out, int B, int Hk, int nsplit) {
    const int bq = blockIdx.x;
    const int Hq = Hk * G;
    const int b = bq / Hq;
    const int qh = bq % Hq;
    const int h = qh / G;
    const int gh = qh % G;
    const int d = threadIdx.x & (DHEAD - 1);
    const int sh = threadIdx.x >> 7;  // 0..3

    extern __shared__ __attribute__((aligned(16))) float sml[];  // [nsplit][2]
    const int base = (b * Hk + h) * nsplit;
    for (int s = threadIdx.x; s < nsplit; s += blockDim.x) {
        const float2 ml = *reinterpret_cast<const float2*>(
            ml_part + ((size_t)(base + s) * G + gh) * 2);
        sml[s * 2] = ml.x;
        sml[s * 2 + 1] = ml.y;
    }
    __syncthreads();

    float mt = -INFINITY;
    for (int s = sh; s < nsplit; s += 4)
        if (sml[s * 2 + 1] > 0.0f) mt = fmaxf(mt, sml[s * 2]);
    float lt = 0.0f, ot = 0.0f;
    for (int s = sh; s < nsplit; s += 4) {
        const float lw = sml[s * 2 + 1];
        if (lw > 0.0f) {
            const float sc = __expf(sml[s * 2] - mt);
            lt += lw * sc;
            ot += o_part[((size_t)(base + s) * G + gh) * DHEAD + d] * sc;
        }
    }
    // merge the 4 split-groups via LDS (per d): (m, l, o) online combine
    __shared__ float red_m[4][DHEAD], red_l[4][DHEAD], red_o[4][DHEAD];
    red_m[sh][d] = mt;
    red_l[sh][d] = lt;
    red_o[sh][d] = ot;
    __syncthreads();
    if (sh == 0) {
#pragma unroll
        for (int g2 = 1; g2 < 4; ++g2) {
            const float m2 = red_m[g2][d], l2 = red_l[g2][d], o2 = red_o[g2][d];
            const float mn = fmaxf(mt, m2);
            const float a1 = (lt > 0.0f) ? __expf(mt - mn) : 0.0f;
            const float a2 = (l2 > 0.0f) ? __expf(m2 - mn) : 0.0f;
            ot = ot * a1 + o2 * a2;
            lt = lt * a1 + l2 * a2;
            mt = (lt > 0.0f) ? mn : -INFINITY;
        }
        const float res = (lt > 0.0f) ? ot / lt : 0.0f;
        reinterpret_cast<uint16_t*>(out)[(size_t)(b * Hq + qh) * DHEAD + d] =
            f32_to_bf16(res);
    }
}

// fused: cnt_ws must be a zero-initialized... no — zeroed HERE each call via
// hipMemsetAsync (graph-capturable memset node). fused=0 keeps the two-kernel
// path (A/B reference and fallback).
extern "C" int oa_attention_decode(void* stream, const void* q, const void* kc,
                                   const void* vc, const void* block_table,
                                   const void* seq_lens, void* o_part,
                                   void* ml_part, void* cnt_ws, void* out, int B,
                                   int Hq, int Hk, int D, int max_blocks,
                                   int block_size, int nsplit, float scale,
                                   int q_stride, int fused) {
    if (q_stride % 8 != 0) return -103;
    const int qs2 = q_stride / 2;
    if (D != DHEAD) return -100;
    if ((block_size & (block_size - 1)) != 0) return -101;
    const int G = Hq / Hk;
    int block_shift = 0;
    while ((1 << block_shift) < block_size) ++block_shift;
    dim3 grid(B * Hk, nsplit), block(256);
    dim3 cgrid(B * Hq), cblock(512);
    const int clds = nsplit * 2 * (int)sizeof(float);

    if (fused) {
        hipError_t e = hipMemsetAsync(cnt_ws, 0, B * Hk * 4, (hipStream_t)stream);
        if (e != hipSuccess) return (int)e;
    }

    const bool deep = B <= 4;
#define LAUNCH_G(GV)                                                                \
    do {                                                                            \
        auto kf = deep ? decode_attn_kernel<GV, true, false, true>                  \
                       : decode_attn_kernel<GV, true, false, false>;                \
        auto ku = deep ? decode_attn_kernel<GV, false, false, true>                 \
                       : decode_attn_kernel<GV, false, false, false>;               \
        if (fused) {                                                                \
            hipLaunchKernelGGL(kf, grid, block,                                     \
                               0, (hipStream_t)stream, (const uint32_t*)q,          \
                               (const uint32_t*)kc, (const uint32_t*)vc,            \
                               (const int*)block_table, (const int*)seq_lens,       \
                               (float*)o_part, (float*)ml_part, (unsigned*)cnt_ws,  \
                               (uint32_t*)out, B, Hk, max_blocks, block_shift,      \
                               nsplit, scale, qs2, (const uint32_t*)nullptr,        \
                               (const uint32_t*)nullptr, (const float*)nullptr,     \
                               (const float*)nullptr, (const int*)nullptr,          \
                               (uint32_t*)nullptr, (uint32_t*)nullptr, 0, 0);       \
            HIP_CHECK_LAUNCH();                                                     \
        } else {                                                                    \
            hipLaunchKernelGGL(ku, grid, block,                                     \
                               0, (hipStream_t)stream, (const uint32_t*)q,          \
                               (const uint32_t*)kc, (const uint32_t*)vc,            \
                               (const int*)block_table, (const int*)seq_lens,       \
                               (float*)o_part, (float*)ml_part, (unsigned*)cnt_ws,  \
                               (uint32_t*)out, B, Hk, max_blocks, block_shift,      \
                               nsplit, scale, qs2, (const uint32_t*)nullptr,        \
                               (const uint32_t*)nullptr, (const float*)nullptr,     \
                               (const float*)nullptr, (const int*)nullptr,          \
                               (uint32_t*)nullptr, (uint32_t*)nullptr, 0, 0);       \
            HIP_CHECK_LAUNCH();                                                     \
            hipLaunchKernelGGL((decode_combine_kernel<GV>), cgrid, cblock, clds,    \
                               (hipStream_t)stream, (const float*)o_part,           \
                               (const float*)ml_part, (uint32_t*)out, B, Hk,        \
                               nsplit);                                             \
            HIP_CHECK_LAUNCH();                                                     \
        }                                                                           \
    } while (0)

    switch (G) {
        case 1: LAUNCH_G(1); break;
        case 2: LAUNCH_G(2); break;
        case 4: LAUNCH_G(4); break;
        case 8: LAUNCH_G(8); break;
        default: return -102;
    }
#undef LAUNCH_G
    return 0;
}

// Fully-fused decode attention: RoPE(q, k) + paged-KV scatter of (k, v) +
// split-K attention in ONE launch (+ the combine kernel) — replaces the
// separate rope_kv launch per layer. q/kin/vin are the RAW (un-roped) GEMV
// outputs; kc/vc are read for cached keys and written at slots[b].
extern "C" int oa_attention_decode_rope(
    void* stream, const void* q, const void* kin, const void* vin, void* kc,
    void* vc, const void* block_table, const void* seq_lens, const void* cos_t,
    const void* sin_t, const void* slots, void* o_part, void* ml_part,
    void* out, int B, int Hq, int Hk, int D, int max_blocks, int block_size,
    int nsplit, float scale, int q_stride, int k_stride, int v_stride) {
    if ((q_stride | k_stride | v_stride) % 8 != 0) return -103;
    if (D != DHEAD) return -100;
    if ((block_size & (block_size - 1)) != 0) return -101;
    const int G = Hq / Hk;
    int block_shift = 0;
    while ((1 << block_shift) < block_size) ++block_shift;
    dim3 grid(B * Hk, nsplit), block(256);
    dim3 cgrid(B * Hq), cblock(512);
    const int clds = nsplit * 2 * (int)sizeof(float);

    const bool deep = B <= 4;
#define LAUNCH_GR(GV)                                                              \
    do {                                                                            \
        auto kr = deep ? decode_attn_kernel<GV, false, true, true>                  \
                       : decode_attn_kernel<GV, false, true, false>;                \
        hipLaunchKernelGGL(kr, grid, block, 0,   \
                           (hipStream_t)stream, (const uint32_t*)q,                 \
                           (const uint32_t*)kc, (const uint32_t*)vc,                \
                           (const int*)block_table, (const int*)seq_lens,           \
                           (float*)o_part, (float*)ml_part, (unsigned*)nullptr,     \
                           (uint32_t*)out, B, Hk, max_blocks, block_shift, nsplit,  \
                           scale, q_stride / 2, (const uint32_t*)kin,               \
                           (const uint32_t*)vin, (const float*)cos_t,               \
                           (const float*)sin_t, (const int*)slots, (uint32_t*)kc,   \
                           (uint32_t*)vc, k_stride / 2, v_stride / 2);              \
        HIP_CHECK_LAUNCH();                                                         \
        hipLaunchKernelGGL((decode_combine_kernel<GV>), cgrid, cblock, clds,        \
                           (hipStream_t)stream, (const float*)o_part,               \
                           (const float*)ml_part, (uint32_t*)out, B, Hk, nsplit);   \
        HIP_CHECK_LAUNCH();                                                         \
    } while (0)

    switch (G) {
        case 1: LAUNCH_GR(1); break;
        case 2: LAUNCH_GR(2); break;
        case 4: LAUNCH_GR(4); break;
        case 8: LAUNCH_GR(8); break;
        default: return -102;
    }
#undef LAUNCH_GR
    return 0;
}
