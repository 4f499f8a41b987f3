// Fused grammar-mask + greedy sampling kernel.
//
// "Constrained-JSON / tool-call sampling kernel (grammar-FSM
// logit mask + greedy/temperature sample on-GPU)" — replaces the reference's
// post-hoc JSON repair (pkg/utils/json.go:16-190) by making invalid JSON
// impossible. The FSM advances on the CPU (C++ — grammar_fsm.cpp); each step
// it produces a per-sequence allowed-token BITMASK (V/32 words, ~16 KB for a
// 128k vocab) which this kernel fuses with the argmax over the logits row —
// one read of the logits, no [B, V] masked-softmax intermediate.
//
// Parallel scheme: grid (chunks, B); each block scans its vocab chunk and
// publishes one packed {ordered-float val, ~idx} u64 via atomicMax (a
// single-block scan of 128k logits was 42 us; 16 chunks bring it near the
// 256-KB read latency). A tiny second kernel decodes the winners.
//
// logits bf16 [B, V]; mask uint32 [B, ceil(V/32)] (bit t set = allowed;
// nullptr -> unconstrained); ws u64 [B] (zeroed here); out int32 [B].

#include "common.h"

__device__ __forceinline__ uint32_t ordered_f32(float f) {
    union {
        uint32_t u;
        float f;
    } v;
    v.f = f;
    return (v.u & 0x80000000u) ? ~v.u : (v.u | 0x80000000u);
}

__global__ __launch_bounds__(256) void masked_argmax_scan(
    const uint32_t* __restrict__ logits, const uint32_t* __restrict__ mask,
    unsigned long long* __restrict__ ws, int V, int mask_words) {
    const int b = blockIdx.y;
    const int chunk = blockIdx.x;
    const int nchunks = gridDim.x;
    const uint32_t* row = logits + (size_t)b * (V / 2);
    const uint32_t* mrow = mask ? mask + (size_t)b * mask_words : nullptr;

    // chunk bounds in word units (8-token granules)
    const int words_total = V / 2;
    const int per_chunk = ((words_total / 4 + nchunks - 1) / nchunks) * 4;
    const int w0 = chunk * per_chunk;
    const int w1 = min(words_total, w0 + per_chunk);

    float best = -INFINITY;
    int besti = -1;
    for (int i = w0 + threadIdx.x * 4; i < w1; i += blockDim.x * 4) {
        uint4 w = *reinterpret_cast<const uint4*>(row + i);
        const int t0 = i * 2;
        uint32_t mbits = 0xffu;
        if (mrow) {
            uint64_t lo = mrow[t0 >> 5];
            uint64_t hi = ((t0 >> 5) + 1 < mask_words) ? mrow[(t0 >> 5) + 1] : 0;
            mbits = (uint32_t)(((lo | (hi << 32)) >> (t0 & 31)) & 0xffu);
        }
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            uint32_t word = (&w.x)[j];
            float lo = bf16_lo(word), hi = bf16_hi(word);
            int tlo = t0 + j * 2, thi = tlo + 1;
            if (((mbits >> (j * 2)) & 1) && (lo > best || (lo == best && tlo < besti))) {
                best = lo;
                besti = tlo;
            }
            if (((mbits >> (j * 2 + 1)) & 1) && (hi > best || (hi == best && thi < besti))) {
                best = hi;
                besti = thi;
            }
        }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        float ov = __shfl_xor(best, off, WAVE);
        int oi = __shfl_xor(besti, off, WAVE);
        if (oi >= 0 && (ov > best || (ov == best && (besti < 0 || oi < besti)))) {
            best = ov;
            besti = oi;
        }
    }
    __shared__ unsigned long long sbest[4];
    const int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0)
        sbest[wid] = besti < 0
                         ? 0ull
                         : ((unsigned long long)ordered_f32(best) << 32) |
                               (unsigned)(0x7fffffff - besti);
    __syncthreads();
    if (threadIdx.x == 0) {
        unsigned long long k = sbest[0];
#pragma unroll
        for (int w = 1; w < 4; ++w) k = max(k, sbest[w]);
        if (k) atomicMax(&ws[b], k);
    }
}

__global__ void masked_argmax_finish(const unsigned long long* __restrict__ ws,
                                     int* __restrict__ out, int B) {
    const int b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b < B) {
        unsigned long long k = ws[b];
        out[b] = k ? (int)(0x7fffffff - (unsigned)(k & 0xffffffffu)) : -1;
    }
}

extern "C" int oa_masked_argmax(void* stream, const void* logits, const void* mask,
                                void* ws, void* out, int B, int V) {
    if (V % 8 != 0) return -100;
    const int mask_words = CEIL_DIV(V, 32);
    hipError_t e = hipMemsetAsync(ws, 0, B * 8, (hipStream_t)stream);
    if (e != hipSuccess) return (int)e;
    const int chunks = 16;
    hipLaunchKernelGGL(masked_argmax_scan, dim3(chunks, B), dim3(256), 0,
                       (hipStream_t)stream, (const uint32_t*)logits,
                       (const uint32_t*)mask, (unsigned long long*)ws, V, mask_words);
    HIP_CHECK_LAUNCH();
    hipLaunchKernelGGL(masked_argmax_finish, dim3(CEIL_DIV(B, 64)), dim3(64), 0,
                       (hipStream_t)stream, (const unsigned long long*)ws, (int*)out, B);
    HIP_CHECK_LAUNCH();
    return 0;
}
