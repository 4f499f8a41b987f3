// Flash-style causal prefill attention for MI355X (gfx950) — SURVEY.md §2b
// "Prefill attention kernel (HIP, bf16 MFMA, LDS-staged KV tiles, gfx950
// tiling)".
//
// v2 structure (close the pipelining gap):
//   * workgroup = 512 threads = 8 waves; wave owns RB x 16 q-rows
//     (RB=2 -> 256-row q-tile; RB=1 -> 128-row tile used when the grid
//     would underfill the 256 CUs at small Sq)
//   * K AND V staged by global_load_lds (16-B pieces, 4 glds per wave per
//     64-key tile) into a 2-deep LDS ring: tile t+1's DMA is in flight
//     under tile t's compute; __syncthreads() doubles as the vmcnt(0)
//     drain (hipcc emits it while a glds is outstanding), which is the
//     guide's verified 2-buffer glds pattern (+40% class vs serial)
//   * K image: 256-B key rows, XOR swizzle byte^=(key&15)<<4 applied on
//     the glds SOURCE address (guide rule 21: the LDS write is
//     lane-linear, so the swizzle moves to the per-lane global address)
//     and on the ds_read_b128 B-fragment reads
//   * V image: [2 key-half][8 d-block] subtiles of [32 key][16 d] bf16
//     with key bits permuted (img_row = (k&3) + ((k>>3)<<2) + ((k&4)<<2))
//     so each ds_read_b64_tr_b16 serves the P·V B-fragment conflict-free:
//     the hardware transpose read replaces v1's 8 scalar ds_write_b16
//     per staged piece (the transpose moves from the write side to a
//     2-cycle read) — guide T10
//   * online softmax per q-row in registers; 16-lane __shfl_xor reductions
//   * P round-trips through a per-wave LDS tile (swizzled) to re-shape
//     C-fragment -> next A-fragment (v3 below REPLACES this with the
//     swapped-QK^T 32x32 in-register form; v2 stays for small grids)
//
// Out-of-range keys are CLAMPED to Skv-1 on the glds source and their
// scores masked to -inf (P=0 nullifies the garbage V contribution), so
// partial tail tiles need no separate path.
//
// Layouts: q [B, Sq, Hq, 128], k/v [B, Skv, Hk, 128] with per-token strides
// (they may be head-slices of one fused qkv buffer), out [B, Sq, Hq, 128]
// contiguous; causal offset = Skv - Sq (query i attends keys [0..Skv-Sq+i])
// so multi-turn chunked prefill reuses the same kernel.

#include "common.h"

#define DHEAD 128
#define KVBLK 64
#define NWAVE 8

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4v;

#define K_BYTES (KVBLK * DHEAD * 2)   // 16 KiB per buffer
#define V_BYTES (KVBLK * DHEAD * 2)   // 16 KiB per buffer

__device__ __forceinline__ uint32_t k_swz_read(int key, int byte_in_row) {
    return (uint32_t)(key * (DHEAD * 2) + (byte_in_row ^ ((key & 15) << 4)));
}

// V image row permutation: key -> subtile row (see header)
__device__ __forceinline__ int v_img_row_inv(int r) {
    // inverse: img_row r -> key
    return (r & 3) + (((r >> 4) & 1) << 2) + (((r >> 2) & 3) << 3);
}

// VRM: stage V ROW-MAJOR (like K — 4 contiguous 256-B rows per DMA chunk,
// ~8 cachelines per glds) instead of the [32 key][16 d] subtile image whose
// gather touches 32 rows x 32 B per chunk (25% line utilization — measured
// DMA-bound at ~11 GB/s/CU). The tr16 reads then address (row, d-quad)
// pieces of the row-major image directly; the same (key&15)<<4 XOR that
// decollides the K reads spreads the tr read's bank pattern.
template <int RB, bool PIPE, bool VRM = false, int KB = KVBLK>
__global__ __launch_bounds__(512, 1) void attn_prefill_v2(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ out, int B, int Hq,
    int Hk, int Sq, int Skv, float scale, int qs, int ks, int vs) {
    constexpr int QROWS = 16 * RB;
    constexpr int QTILE = QROWS * NWAVE;
    constexpr int P_BYTES = QROWS * KB * 2;
    constexpr int KBYTES = KB * DHEAD * 2;   // one K (or V) ring buffer
    constexpr int PIECES = KB / 32;          // 1-KiB DMA chunks per wave
    constexpr int NKB = KB / 16;             // 16-key QK column blocks
    constexpr int NKB2 = KB / 32;            // 32-key PV k-blocks
    __shared__ __attribute__((aligned(16))) char smem[4 * KBYTES +
                                                      NWAVE * P_BYTES];
    const uint32_t voff = 2 * KBYTES;        // V buffers after the K ring
    const uint32_t poff = voff + 2 * KBYTES;

    const int qtile = blockIdx.x;
    const int bh = blockIdx.y;
    const int b = bh / Hq;
    const int h = bh % Hq;
    const int hk = h / (Hq / Hk);
    const int offset = Skv - Sq;  // causal diagonal offset

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int fr = lane & 15;  // fragment row/col index (0..15)
    const int fs = lane >> 4;  // k-slice 0..3

    // ---- Q fragments: aq[rb][dblk] = Q[qrow = rb*16 + fr][d = dblk*32 + fs*8]
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

    float m[RB][4], lsum[RB][4];
    f32x4_t o_acc[RB][8];
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

    const int kv_needed = min(Skv, offset + qtile * QTILE + QTILE);
    const int ntiles = CEIL_DIV(max(kv_needed, 0), KB);

    const char* kbase = reinterpret_cast<const char*>(k) +
                        ((size_t)b * Skv * ks + (size_t)hk * DHEAD) * 2;
    const char* vbase = reinterpret_cast<const char*>(v) +
                        ((size_t)b * Skv * vs + (size_t)hk * DHEAD) * 2;

    const int skv_clamp = Skv - 1;

    // Each wave issues 2 K-glds + 2 V-glds per tile (chunk c = wid*2 + p):
    //   K chunk covers keys c*4 + (lane>>4), XOR-swizzled source address
    //   V chunk is subtile c: lane -> image row lane>>1 (permuted keys)
    //
    // The DMA is issued by INLINE ASM, not the builtin: hipcc's machine-
    // level wait inserter cannot disambiguate a runtime LDS-DMA destination
    // from the compute phase's ds_reads and drains the ring with a vmcnt(0)
    // before the first ds_read of every tile — which serializes the whole
    // pipeline (measured in the .s for both builtin forms). Untracked asm
    // glds leave wait placement to us: one explicit vmcnt(0) right before
    // each tile barrier, so tile t+1's DMA streams under tile t's MFMAs.
    // (Untracked VMEM only ever makes the compiler's own counted waits
    // MORE conservative — vmcnt retires in issue order — never unsafe.)
    //
    // EVERY asm operand is a LOOP-PERSISTENT value updated only at the loop
    // top: the wait inserter marks asm operands as pending defs, so if a
    // compute-phase instruction reuses one of those registers it inserts a
    // drain mid-compute (measured: vmcnt(3..0) before the first MFMAs when
    // the address temps died into ds_read destinations). Persistent
    // operands pin the registers for the whole loop; the only inserted
    // waits land right after our own vmcnt(0), where they are free.

    // per-piece invariants
    int k_byte[PIECES], v_byte[PIECES], kkey0[PIECES], vkey0[PIECES];
#pragma unroll
    for (int p = 0; p < PIECES; ++p) {
        const int c = wid * PIECES + p;
        const int kkey = c * 4 + (lane >> 4);
        kkey0[p] = kkey;
        k_byte[p] = ((lane & 15) * 16) ^ ((kkey & 15) << 4);
        if (VRM) {
            vkey0[p] = kkey;
            v_byte[p] = k_byte[p];
        } else {
            const int kb2 = c >> 3, nb = c & 7;
            vkey0[p] = kb2 * 32 + v_img_row_inv(lane >> 1);
            v_byte[p] = (nb * 16 + (lane & 1) * 8) * 2;
        }
    }
    // LDS destinations (uniform per wave): [buf][piece] for K and V
    uint32_t m0k[2][PIECES], m0v[2][PIECES];
#pragma unroll
    for (int buf = 0; buf < 2; ++buf)
#pragma unroll
        for (int p = 0; p < PIECES; ++p) {
            const int c = wid * PIECES + p;
            m0k[buf][p] = __builtin_amdgcn_readfirstlane(
                (uint32_t)(uintptr_t)smem + buf * KBYTES + c * 1024);
            m0v[buf][p] = __builtin_amdgcn_readfirstlane(
                (uint32_t)(uintptr_t)smem + voff + buf * KBYTES + c * 1024);
        }
    // persistent asm operands: current tile's source addresses + LDS bases
    const char* ksrc[PIECES];
    const char* vsrc[PIECES];
    uint32_t m0k_cur[PIECES], m0v_cur[PIECES];
    auto set_tile = [&](int t, int buf) {
        const int kv0 = t * KB;
#pragma unroll
        for (int p = 0; p < PIECES; ++p) {
            ksrc[p] = kbase + (size_t)min(kv0 + kkey0[p], skv_clamp) * ks * 2 +
                      k_byte[p];
            vsrc[p] = vbase + (size_t)min(kv0 + vkey0[p], skv_clamp) * vs * 2 +
                      v_byte[p];
            m0k_cur[p] = m0k[buf][p];
            m0v_cur[p] = m0v[buf][p];
        }
    };
    auto stage = [&]() {
#pragma unroll
        for (int p = 0; p < PIECES; ++p) {
            asm volatile(
                "s_mov_b32 m0, %0\n\t"
                "global_load_lds_dwordx4 %1, off"
                :
                : "s"(m0k_cur[p]), "v"(ksrc[p]));
            asm volatile(
                "s_mov_b32 m0, %0\n\t"
                "global_load_lds_dwordx4 %1, off"
                :
                : "s"(m0v_cur[p]), "v"(vsrc[p]));
        }
    };

    auto compute = [&](int t, const char* kbuf, const char* vbuf) {
        const int kv0 = t * KB;

#pragma unroll
        for (int rb = 0; rb < RB; ++rb) {
            // ---- QK^T ----------------------------------------------------
            f32x4_t s[NKB];
#pragma unroll
            for (int kb = 0; kb < NKB; ++kb) {
                f32x4_t acc = (f32x4_t){0.f, 0.f, 0.f, 0.f};
#pragma unroll
                for (int dblk = 0; dblk < 4; ++dblk) {
                    uint4 bk = *reinterpret_cast<const uint4*>(
                        kbuf + k_swz_read(kb * 16 + fr, (dblk * 32 + fs * 8) * 2));
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
            for (int kb = 0; kb < NKB; ++kb) {
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
            // fully-masked padding rows produce NaN locally, never stored.

#pragma unroll
            for (int kb = 0; kb < NKB; ++kb) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const float p = __expf(s[kb][r] - m[rb][r]);
                    s[kb][r] = p;
                    psum[r] += p;
                    *reinterpret_cast<uint16_t*>(
                        smem + poff + wid * P_BYTES +
                        (rb * 16 + fs * 4 + r) * (KB * 2) +
                        (((kb * 16 + fr) * 2) ^ (((rb * 16 + fs * 4 + r) & 7) << 4))) =
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

            // ---- P·V (V via hardware transpose reads) --------------------
            uint4 ap[NKB2];
#pragma unroll
            for (int kb2 = 0; kb2 < NKB2; ++kb2) {
                const int prow = rb * 16 + fr;
                ap[kb2] = *reinterpret_cast<const uint4*>(
                    smem + poff + wid * P_BYTES + prow * (KB * 2) +
                    (((kb2 * 32 + fs * 8) * 2) ^ ((prow & 7) << 4)));
            }
            // VRM per-lane supplier addressing: lane i (in its 16-lane
            // group) supplies piece (row i>>2, d-quad i&3) of the [4 key]
            // x[16 d] block; the hardware transpose hands lane i column i.
            const int vq8 = (fr & 3) * 8;
            const int vrow0 = fs * 8 + (fr >> 2);       // read0 key-in-32
            const int vrow1 = vrow0 + 4;                // read1
#pragma unroll
            for (int nb = 0; nb < 8; ++nb) {
#pragma unroll
                for (int kb2 = 0; kb2 < NKB2; ++kb2) {
                    bf16x4v r0, r1;
                    if (VRM) {
                        const char* vb2 = vbuf + kb2 * 8192;
                        const int t0 = ((nb * 32) | vq8) ^ ((vrow0 & 15) << 4);
                        const int t1 = ((nb * 32) | vq8) ^ ((vrow1 & 15) << 4);
                        r0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                            (__attribute__((address_space(3))) bf16x4v*)(
                                vb2 + vrow0 * 256 + t0));
                        r1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                            (__attribute__((address_space(3))) bf16x4v*)(
                                vb2 + vrow1 * 256 + t1));
                    } else {
                        const char* sub = vbuf + (kb2 * 8 + nb) * 1024;
                        r0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                            (__attribute__((address_space(3))) bf16x4v*)(sub + lane * 8));
                        r1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                            (__attribute__((address_space(3))) bf16x4v*)(sub + 512 + lane * 8));
                    }
                    union {
                        struct { bf16x4v lo, hi; } p;
                        bf16x8_t v8;
                    } bv;
                    bv.p.lo = r0;
                    bv.p.hi = r1;
                    o_acc[rb][nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        *reinterpret_cast<bf16x8_t*>(&ap[kb2]), bv.v8,
                        o_acc[rb][nb], 0, 0, 0);
                }
            }
        }
    };

    // 2-deep ring, hand-unrolled so every buffer index is a literal: tile
    // t+1's glds stream under tile t's MFMAs; __syncthreads() both drains
    // this wave's own glds (hipcc emits the vmcnt(0) there) and fences the
    // buffer swap across waves.
    // Drain + keep the staging addresses LIVE across compute: without the
    // dummy operands the address registers die at the glds asm, the
    // allocator reuses them for ds_read destinations, and the wait inserter
    // (which models asm operands as pending) re-serializes the pipeline
    // with vmcnt(3..0) in front of the first MFMAs (measured).
    auto drain = [&]() {
        if constexpr (PIECES == 2) {
            asm volatile("s_waitcnt vmcnt(0)"
                         :
                         : "v"(ksrc[0]), "v"(ksrc[1]), "v"(vsrc[0]),
                           "v"(vsrc[1]), "s"(m0k_cur[0]), "s"(m0k_cur[1]),
                           "s"(m0v_cur[0]), "s"(m0v_cur[1])
                         : "memory");
        } else {
            static_assert(PIECES == 4, "drain operand list covers 2/4 pieces");
            asm volatile("s_waitcnt vmcnt(0)"
                         :
                         : "v"(ksrc[0]), "v"(ksrc[1]), "v"(ksrc[2]),
                           "v"(ksrc[3]), "v"(vsrc[0]), "v"(vsrc[1]),
                           "v"(vsrc[2]), "v"(vsrc[3]), "s"(m0k_cur[0]),
                           "s"(m0k_cur[1]), "s"(m0k_cur[2]), "s"(m0k_cur[3]),
                           "s"(m0v_cur[0]), "s"(m0v_cur[1]), "s"(m0v_cur[2]),
                           "s"(m0v_cur[3])
                         : "memory");
        }
    };
    if (ntiles > 0) {
        set_tile(0, 0);
        stage();
    }
    // static priority for the second-dispatched half (guide T5 static
    // form): the younger 4 waves lose VALU arbitration on every segment;
    // one setprio before the loop removes their start-of-segment penalty.
    // The readfirstlane guard keeps the condition provably wave-uniform
    // (a divergent guard lowers to exec-masking, which s_setprio ignores).
    if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
        __builtin_amdgcn_s_setprio(1);
    drain();
    __syncthreads();
    for (int t = 0; t < ntiles; ++t) {
        const int cur = t & 1;
        if (t + 1 < ntiles) set_tile(t + 1, cur ^ 1);
        if (PIPE && t + 1 < ntiles) stage();
        compute(t, smem + cur * KBYTES, smem + voff + cur * KBYTES);
        if (!PIPE && t + 1 < ntiles) stage();  // A/B reference: issue late
        drain();          // tile t+1's DMA landed
        __syncthreads();  // all waves' reads of buf[cur] complete
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

// ---------------------------------------------------------------------------
// v3: swapped-QK^T structure on 32x32x16 MFMAs with IN-REGISTER P (guide T12)
//
//   * wave owns 32 q-rows; lane's l&31 IS its q-row, so softmax stats are
//     per-lane scalars and each row reduction is ONE __shfl_xor(.,32)
//     (v2: 4-step group16 reductions per row-quad)
//   * S^T = mfma(A=K, B=Q): C[row=key][col=q-row] puts all 64 of a lane's
//     key-scores (of its row; partner lane l^32 holds the other 64) in the
//     four 32x32 accumulators — masked/softmaxed in registers
//   * P -> PV A-operand WITHOUT the LDS round-trip: pack score quads to
//     bf16 pairs, exchange the partner half with __shfl_xor(.,32), and feed
//     the assembled 8-key fragments straight into mfma(A=V^T, B=P^T) so
//     O[row=dim][col=q-row] keeps the lane<->q-row mapping (alpha rescale
//     stays a per-lane scalar multiply)
//   * K/V LDS images, glds staging ring, drain discipline: identical to v2
//     (KB=128); the per-wave P tile is GONE (128 KiB LDS total)
//   * per-tile wave-uniform branches skip masking on fully-unmasked tiles
//     and skip compute entirely on tiles past the wave's causal diagonal
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(16))) float f32x16_t;

// Partner-half (lane ^ 32) value on the VALU: v_permlane32_swap with equal
// inputs yields r[0] = [lo|lo], r[1] = [hi|hi], so the partner's value is
// r[1] for the low half and r[0] for the high half. The two-asm-operand
// form is NOT equivalent (measured on-box: the upper half saw its own
// value) — the builtin models both register updates correctly.
__device__ __forceinline__ uint32_t xor32_u32(uint32_t v, int lh) {
    auto r = __builtin_amdgcn_permlane32_swap(v, v, false, false);
    return lh ? r[0] : r[1];
}
__device__ __forceinline__ float xor32_f32(float v, int lh) {
    return __uint_as_float(xor32_u32(__float_as_uint(v), lh));
}

// GQ > 1 (v4): the workgroup carries GQ q-heads that SHARE one KV head
// (requires Hq/Hk == GQ) — waves split into GQ groups of NWAVE/GQ, each
// group owning QTILE/GQ q-rows of its head. Every staged K/V tile feeds
// all GQ heads, cutting the kernel's total KV DMA by GQ: the v3 PMC put
// the residual WAIT at the staged-KV demand of the co-resident
// workgroups (~9.6 TB/s against L2+HBM), which this divides by 4.
template <int KB = 128, int GQ = 1>
__global__ __launch_bounds__(512, 1) void attn_prefill_v3(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ out, int B, int Hq,
    int Hk, int Sq, int Skv, float scale, int qs, int ks, int vs) {
    constexpr int QROWS = 32;                // per wave
    constexpr int QTILE = QROWS * NWAVE;     // 256 across the WG
    constexpr int QTILE_H = QTILE / GQ;      // rows per head per WG
    constexpr int KBYTES = KB * DHEAD * 2;
    constexpr int PIECES = KB / 32;          // 1-KiB DMA chunks per wave
    constexpr int NKB32 = KB / 32;           // 32-key score blocks
    __shared__ __attribute__((aligned(16))) char smem[4 * KBYTES];
    const uint32_t voff = 2 * KBYTES;

    const int qtile = blockIdx.x;
    const int bh = blockIdx.y;               // spans B * (Hq/GQ)
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int b = bh / (Hq / GQ);
    const int h = (bh % (Hq / GQ)) * GQ + wid / (NWAVE / GQ);
    const int hk = h / (Hq / Hk);
    const int offset = Skv - Sq;

    const int l31 = lane & 31;  // the lane's q-row within the wave block
    const int lh = lane >> 5;   // fragment k-half (and score key-quad offset)

    // ---- Q: B-fragment per d-step s: bq[s] = Q[row l31][d = s*16 + lh*8 ..+7]
    const int qrow0 = qtile * QTILE_H + (wid % (NWAVE / GQ)) * QROWS;
    const int qrow = qrow0 + l31;
    uint4 bq[8];
    if (qrow < Sq) {
        const uint16_t* qp = q + (size_t)(b * Sq + qrow) * qs + (size_t)h * DHEAD;
#pragma unroll
        for (int s = 0; s < 8; ++s)
            bq[s] = *reinterpret_cast<const uint4*>(qp + s * 16 + lh * 8);
    } else {
#pragma unroll
        for (int s = 0; s < 8; ++s) bq[s] = make_uint4(0, 0, 0, 0);
    }

    float m = -INFINITY, lsum = 0.0f;
    f32x16_t oa[4];
#pragma unroll
    for (int db = 0; db < 4; ++db) oa[db] = (f32x16_t)(0.0f);

    const int kv_needed = min(Skv, offset + qtile * QTILE_H + QTILE_H);
    const int ntiles = CEIL_DIV(max(kv_needed, 0), KB);

    const char* kbase = reinterpret_cast<const char*>(k) +
                        ((size_t)b * Skv * ks + (size_t)hk * DHEAD) * 2;
    const char* vbase = reinterpret_cast<const char*>(v) +
                        ((size_t)b * Skv * vs + (size_t)hk * DHEAD) * 2;
    const int skv_clamp = Skv - 1;

    // glds staging: identical images and discipline to v2 (see the comment
    // blocks above attn_prefill_v2) — K row-major XOR-swizzled, V subtile
    // image for the tr16 B-fragments, inline-asm DMA with loop-persistent
    // operands and an explicit vmcnt(0) drain.
    int k_byte[PIECES], v_byte[PIECES], kkey0[PIECES], vkey0[PIECES];
#pragma unroll
    for (int p = 0; p < PIECES; ++p) {
        const int c = wid * PIECES + p;
        const int kkey = c * 4 + (lane >> 4);
        kkey0[p] = kkey;
        k_byte[p] = ((lane & 15) * 16) ^ ((kkey & 15) << 4);
        const int kb2 = c >> 3, nb = c & 7;
        vkey0[p] = kb2 * 32 + v_img_row_inv(lane >> 1);
        v_byte[p] = (nb * 16 + (lane & 1) * 8) * 2;
    }
    uint32_t m0k[2][PIECES], m0v[2][PIECES];
#pragma unroll
    for (int buf = 0; buf < 2; ++buf)
#pragma unroll
        for (int p = 0; p < PIECES; ++p) {
            const int c = wid * PIECES + p;
            m0k[buf][p] = __builtin_amdgcn_readfirstlane(
                (uint32_t)(uintptr_t)smem + buf * KBYTES + c * 1024);
            m0v[buf][p] = __builtin_amdgcn_readfirstlane(
                (uint32_t)(uintptr_t)smem + voff + buf * KBYTES + c * 1024);
        }
    const char* ksrc[PIECES];
    const char* vsrc[PIECES];
    uint32_t m0k_cur[PIECES], m0v_cur[PIECES];
    auto set_tile = [&](int t, int buf) {
        const int kv0 = t * KB;
#pragma unroll
        for (int p = 0; p < PIECES; ++p) {
            ksrc[p] = kbase + (size_t)min(kv0 + kkey0[p], skv_clamp) * ks * 2 +
                      k_byte[p];
            vsrc[p] = vbase + (size_t)min(kv0 + vkey0[p], skv_clamp) * vs * 2 +
                      v_byte[p];
            m0k_cur[p] = m0k[buf][p];
            m0v_cur[p] = m0v[buf][p];
        }
    };
    auto stage = [&]() {
#pragma unroll
        for (int p = 0; p < PIECES; ++p) {
            asm volatile(
                "s_mov_b32 m0, %0\n\t"
                "global_load_lds_dwordx4 %1, off"
                :
                : "s"(m0k_cur[p]), "v"(ksrc[p]));
            asm volatile(
                "s_mov_b32 m0, %0\n\t"
                "global_load_lds_dwordx4 %1, off"
                :
                : "s"(m0v_cur[p]), "v"(vsrc[p]));
        }
    };
    auto drain = [&]() {
        static_assert(PIECES == 4, "drain operand list covers KB=128");
        asm volatile("s_waitcnt vmcnt(0)"
                     :
                     : "v"(ksrc[0]), "v"(ksrc[1]), "v"(ksrc[2]), "v"(ksrc[3]),
                       "v"(vsrc[0]), "v"(vsrc[1]), "v"(vsrc[2]), "v"(vsrc[3]),
                       "s"(m0k_cur[0]), "s"(m0k_cur[1]), "s"(m0k_cur[2]),
                       "s"(m0k_cur[3]), "s"(m0v_cur[0]), "s"(m0v_cur[1]),
                       "s"(m0v_cur[2]), "s"(m0v_cur[3])
                     : "memory");
    };

    const int qpos = offset + qrow;                    // per-lane diagonal
    const int wave_kv_last = offset + qrow0 + QROWS;   // first key PAST the
                                                       // wave's last diagonal

    auto compute = [&](int t, const char* kbuf, const char* vbuf,
                       bool need_mask) {
        const int kv0 = t * KB;

        // ---- S^T = K·Q^T: four 32x32 blocks of 32 keys ---------------------
        f32x16_t sa[NKB32];
#pragma unroll
        for (int kb = 0; kb < NKB32; ++kb) sa[kb] = (f32x16_t)(0.0f);
#pragma unroll
        for (int s = 0; s < 8; ++s)
#pragma unroll
            for (int kb = 0; kb < NKB32; ++kb) {
                uint4 ak = *reinterpret_cast<const uint4*>(
                    kbuf + k_swz_read(kb * 32 + l31, s * 32 + lh * 16));
                sa[kb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    *reinterpret_cast<bf16x8_t*>(&ak),
                    *reinterpret_cast<bf16x8_t*>(&bq[s]), sa[kb], 0, 0, 0);
            }

        // ---- mask + online softmax (per-lane row, RAW-score domain) --------
        // lane's score i of block kb = key kv0 + kb*32 + (i&3) + 8*(i>>2) + 4*lh
        // Scores stay UNSCALED: scale*log2(e) folds into the exp2 argument
        // (one v_fma per score instead of a separate scale pass + expf's
        // internal log2e multiply — v_exp_f32 IS exp2).
        const float scale2 = scale * 1.4426950408889634f;
        if (need_mask) {  // wave-uniform: one branch, not per-score selects.
            // Fold the tile-varying part into two per-lane values so each
            // score masks with compares against inline constants (keeping
            // the 64 kg values out of SGPRs — they spilled as writelanes).
            const int dqk = 4 * lh + kv0 - qpos;   // mask if dqk + c > 0
            const int dskv = 4 * lh + kv0 - Skv;   // mask if dskv + c >= 0
#pragma unroll
            for (int kb = 0; kb < NKB32; ++kb)
#pragma unroll
                for (int i = 0; i < 16; ++i) {
                    const int c = kb * 32 + (i & 3) + 8 * (i >> 2);
                    if (dqk > -c || dskv >= -c) sa[kb][i] = -INFINITY;
                }
        }
        float rowmax = -INFINITY;
#pragma unroll
        for (int kb = 0; kb < NKB32; ++kb)
#pragma unroll
            for (int i = 0; i < 16; ++i) rowmax = fmaxf(rowmax, sa[kb][i]);
        // half-wave exchange on the VALU (v_permlane32_swap), not ds_bpermute:
        // the LDS pipe stays free for the b128/tr16 operand reads
        rowmax = fmaxf(rowmax, xor32_f32(rowmax, lh));

        const float m_old = m;
        const float mn = fmaxf(m_old, rowmax);
        m = mn;
        const float mnb = mn * scale2;
        float psum = 0.0f;

        // ---- exp + pack to bf16 quads: pk[kb][g] = keys kb*32+g*8+4*lh+{0..3}
        uint32_t pk[NKB32][4][2];
#pragma unroll
        for (int kb = 0; kb < NKB32; ++kb)
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                float p0 = __builtin_amdgcn_exp2f(
                    __builtin_fmaf(sa[kb][g * 4 + 0], scale2, -mnb));
                float p1 = __builtin_amdgcn_exp2f(
                    __builtin_fmaf(sa[kb][g * 4 + 1], scale2, -mnb));
                float p2 = __builtin_amdgcn_exp2f(
                    __builtin_fmaf(sa[kb][g * 4 + 2], scale2, -mnb));
                float p3 = __builtin_amdgcn_exp2f(
                    __builtin_fmaf(sa[kb][g * 4 + 3], scale2, -mnb));
                psum += (p0 + p1) + (p2 + p3);
                asm("v_cvt_pk_bf16_f32 %0, %1, %2"
                    : "=v"(pk[kb][g][0]) : "v"(p0), "v"(p1));
                asm("v_cvt_pk_bf16_f32 %0, %1, %2"
                    : "=v"(pk[kb][g][1]) : "v"(p2), "v"(p3));
            }
        psum += xor32_f32(psum, lh);
        // rescale only when some lane's running max moved (wave vote): after
        // the first few tiles most tiles leave every row max unchanged and
        // the 64 accumulator multiplies + lsum scale are skipped outright
        if (__any(rowmax > m_old)) {
            const float alpha = __builtin_amdgcn_exp2f((m_old - mn) * scale2);
            lsum *= alpha;
#pragma unroll
            for (int db = 0; db < 4; ++db)
#pragma unroll
                for (int i = 0; i < 16; ++i) oa[db][i] *= alpha;
        }
        lsum += psum;

        // ---- P·V: O^T[dim][q-row] via mfma(A=V^T, B=P^T) -------------------
        // k-step ks2 covers keys ks2*16 + lh*8 + {0..7}: own quad = pk[kb][e*2
        // + lh], partner quad arrives by shfl_xor(32) of pk[kb][e*2 + (lh^1)]
        // (each lane sends what its partner needs and receives what it needs).
#pragma unroll
        for (int ks2 = 0; ks2 < KB / 16; ++ks2) {
            const int kb = ks2 >> 1, e2 = (ks2 & 1) * 2;
            const uint32_t own0 = lh ? pk[kb][e2 + 1][0] : pk[kb][e2][0];
            const uint32_t own1 = lh ? pk[kb][e2 + 1][1] : pk[kb][e2][1];
            const uint32_t snd0 = lh ? pk[kb][e2][0] : pk[kb][e2 + 1][0];
            const uint32_t snd1 = lh ? pk[kb][e2][1] : pk[kb][e2 + 1][1];
            const uint32_t rcv0 = xor32_u32(snd0, lh);
            const uint32_t rcv1 = xor32_u32(snd1, lh);
            union {
                uint32_t u[4];
                bf16x8_t v8;
            } pf;
            pf.u[0] = lh ? rcv0 : own0;
            pf.u[1] = lh ? rcv1 : own1;
            pf.u[2] = lh ? own0 : rcv0;
            pf.u[3] = lh ? own1 : rcv1;

            // V^T A-fragment: subtile tr16 reads — group-uniform 128-B runs
            // (fs_eff = (ks2&1)*2 + lh selects the key octet, col = l31&15)
            const int sub_row = ((ks2 & 1) * 2 + lh) * 16 + (l31 & 15);
#pragma unroll
            for (int db = 0; db < 4; ++db) {
                const char* sub =
                    vbuf + ((kb * 8) + db * 2 + (l31 >> 4)) * 1024;
                bf16x4v r0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                    (__attribute__((address_space(3))) bf16x4v*)(sub +
                                                                 sub_row * 8));
                bf16x4v r1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                    (__attribute__((address_space(3))) bf16x4v*)(sub + 512 +
                                                                 sub_row * 8));
                union {
                    struct { bf16x4v lo, hi; } p;
                    bf16x8_t v8;
                } bv;
                bv.p.lo = r0;
                bv.p.hi = r1;
                oa[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    bv.v8, pf.v8, oa[db], 0, 0, 0);
            }
        }
    };

    if (ntiles > 0) {
        set_tile(0, 0);
        stage();
    }
    if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
        __builtin_amdgcn_s_setprio(1);
    drain();
    __syncthreads();
    for (int t = 0; t < ntiles; ++t) {
        const int cur = t & 1;
        const int kv0 = t * KB;
        if (t + 1 < ntiles) set_tile(t + 1, cur ^ 1);
        if (t + 1 < ntiles) stage();
        if (kv0 < wave_kv_last) {  // wave-uniform: skip past-diagonal tiles
            // fully-unmasked tile: every key <= every lane's diagonal
            const bool need_mask =
                !(kv0 + KB <= offset + qrow0 + 1 && kv0 + KB <= Skv);
            compute(t, smem + cur * KBYTES, smem + voff + cur * KBYTES,
                    need_mask);
        }
        drain();
        __syncthreads();
    }

    // ---- epilogue: out[q-row][d] = o/l; lane holds dims (i&3)+8*(i>>2)+4*lh
    if (qrow < Sq) {
        const float inv_l = (lsum > 0.0f) ? 1.0f / lsum : 0.0f;
        uint16_t* op = out + ((size_t)(b * Sq + qrow) * Hq + h) * DHEAD;
#pragma unroll
        for (int db = 0; db < 4; ++db)
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                const int d0 = db * 32 + g * 8 + 4 * lh;
                uint2 pkd;
                pkd.x = (uint32_t)f32_to_bf16(oa[db][g * 4 + 0] * inv_l) |
                        ((uint32_t)f32_to_bf16(oa[db][g * 4 + 1] * inv_l) << 16);
                pkd.y = (uint32_t)f32_to_bf16(oa[db][g * 4 + 2] * inv_l) |
                        ((uint32_t)f32_to_bf16(oa[db][g * 4 + 3] * inv_l) << 16);
                *reinterpret_cast<uint2*>(op + d0) = pkd;
            }
    }
}

extern "C" int oa_attention_prefill_variant(
    void* stream, const void* q, const void* k, const void* v, void* out,
    int B, int Hq, int Hk, int Sq, int Skv, int D, float scale,
    int q_stride, int k_stride, int v_stride, int variant) {
    if (D != DHEAD) return -100;
    if (Hq % Hk != 0) return -101;
    if ((q_stride | k_stride | v_stride) % 8 != 0) return -102;
    dim3 block(512);
    const dim3 grid1(CEIL_DIV(Sq, 128), B * Hq);
    const dim3 grid2(CEIL_DIV(Sq, 256), B * Hq);
#define LAUNCH(KERN, GRID)                                                      \
    hipLaunchKernelGGL(KERN, GRID, block, 0, (hipStream_t)stream,               \
                       (const uint16_t*)q, (const uint16_t*)k,                  \
                       (const uint16_t*)v, (uint16_t*)out, B, Hq, Hk, Sq, Skv,  \
                       scale, q_stride, k_stride, v_stride)
    // RB=2 (256-row q-tile) instantiations were REMOVED: every form spilled
    // 25-43 VGPRs past the 256 cap and lost all A/Bs to the RB1 kernels.
    (void)grid2;
    switch (variant) {
        case 1:  // RB1 pipelined, 64-key tiles
            LAUNCH((attn_prefill_v2<1, true>), grid1);
            break;
        case 3:  // RB1 late-issue (A/B reference: DMA after compute)
            LAUNCH((attn_prefill_v2<1, false>), grid1);
            break;
        case 5:  // RB1 pipelined, row-major V (coalesced V DMA — measured
                 // 15% SLOWER than the subtile image; kept for A/B)
            LAUNCH((attn_prefill_v2<1, true, true>), grid1);
            break;
        case 7:  // RB1 pipelined, 128-key tiles (160 KiB LDS: half the
                 // barriers per key, double the MFMAs per phase)
            LAUNCH((attn_prefill_v2<1, true, false, 128>), grid1);
            break;
        case 8:  // v3: swapped-QK^T 32x32 MFMAs, in-register P (256-row tile)
            LAUNCH((attn_prefill_v3<128>), dim3(CEIL_DIV(Sq, 256), B * Hq));
            break;
        case 9:  // v4: GQA-merged v3 — GQ q-heads share each staged KV tile
            if (Hq == 2 * Hk)
                LAUNCH((attn_prefill_v3<128, 2>),
                       dim3(CEIL_DIV(Sq, 128), B * (Hq / 2)));
            else if (Hq == 4 * Hk)
                LAUNCH((attn_prefill_v3<128, 4>),
                       dim3(CEIL_DIV(Sq, 64), B * (Hq / 4)));
            else if (Hq == 8 * Hk)
                LAUNCH((attn_prefill_v3<128, 8>),
                       dim3(CEIL_DIV(Sq, 32), B * (Hq / 8)));
            else
                return -104;
            break;
        default:
            return -103;
    }
#undef LAUNCH
    HIP_CHECK_LAUNCH();
    return 0;
}

extern "C" int oa_attention_prefill(void* stream, const void* q, const void* k,
                                    const void* v, void* out, int B, int Hq,
                                    int Hk, int Sq, int Skv, int D, float scale,
                                    int q_stride, int k_stride, int v_stride) {
    // OPSAGENT_PREFILL_VARIANT=1..4 forces a variant (read per call so one
    // process can A/B all variants).
    const char* e = getenv("OPSAGENT_PREFILL_VARIANT");
    int variant = e ? atoi(e) : 0;
    // A/B-measured on MI355X (profiles/README.md): the GQA-merged form
    // (v4) shares each staged KV tile across the q-heads of one KV head
    // and wins whenever it applies (S8192 721 vs 556 vs 350 TF; S2048 439;
    // B16/S1024 426): the v3 PMC put the residual WAIT at staged-KV DMA
    // pressure, which merging divides by GQ. v3 covers non-4/8 GQA
    // ratios; below ~256 workgroups the 128-row v2 keeps more CUs busy.
    if (variant <= 0) {
        const int gq = (Hk > 0 && Hq % Hk == 0) ? Hq / Hk : 1;
        const int64_t grid256 = (int64_t)CEIL_DIV(Sq, 256) * B * Hq;
        if ((gq == 2 || gq == 4 || gq == 8) && grid256 >= 256)
            variant = 9;
        else if (grid256 >= 256)
            variant = 8;
        else
            variant = 7;
    }
    return oa_attention_prefill_variant(stream, q, k, v, out, B, Hq, Hk, Sq,
                                        Skv, D, scale, q_stride, k_stride,
                                        v_stride, variant);
}
