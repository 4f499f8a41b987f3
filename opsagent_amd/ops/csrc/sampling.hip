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
// logits bf16 [B, V]; mask uint32 [B, ceil(V/32)] (bit t set = token allowed;
// mask == nullptr -> unconstrained); out int32 [B]. Greedy (temperature ~0 is
// the agent's sampling mode, ref openai.go:74); temperature sampling runs via
// gumbel noise pre-added by the host when requested.

#include "common.h"

__global__ __launch_bounds__(256) void masked_argmax_kernel(
    const uint32_t* __restrict__ logits,  // [B, V/2] packed bf16x2
    const uint32_t* __restrict__ mask,    // [B, mask_words] or nullptr
    int* __restrict__ out, int V, int mask_words) {
    const int b = blockIdx.x;
    const uint32_t* row = logits + (size_t)b * (V / 2);
    const uint32_t* mrow = mask ? mask + (size_t)b * mask_words : nullptr;

    float best = -INFINITY;
    int besti = -1;
    // 8 tokens (4 words) per iteration per lane
    for (int i = threadIdx.x * 4; i < V / 2; i += blockDim.x * 4) {
        uint4 w = *reinterpret_cast<const uint4*>(row + i);
        const int t0 = i * 2;
        uint32_t mbits = 0xffffffffu;
        if (mrow) {
            // 8 consecutive tokens starting at t0: within one 32-bit mask word
            // when t0 % 32 <= 24; t0 is a multiple of 8 so they span at most
            // one word boundary; handle generally with a 64-bit window.
            uint64_t lo = mrow[t0 >> 5];
            uint64_t hi = ((t0 >> 5) + 1 < mask_words) ? mrow[(t0 >> 5) + 1] : 0;
            mbits = (uint32_t)(((lo | (hi << 32)) >> (t0 & 31)) & 0xffu);
        }
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            uint32_t word = (&w.x)[j];
            float lo = bf16_lo(word), hi = bf16_hi(word);
            int tlo = t0 + j * 2, thi = t0 + j * 2 + 1;
            bool alo = (mbits >> (j * 2)) & 1, ahi = (mbits >> (j * 2 + 1)) & 1;
            // deterministic tie-break: lowest index wins
            if (alo && (lo > best || (lo == best && tlo < besti))) { best = lo; besti = tlo; }
            if (ahi && (hi > best || (hi == best && thi < besti))) { best = hi; besti = thi; }
        }
    }
    // wave reduce (value, index) — prefer higher value, then lower index
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        float ov = __shfl_xor(best, off, WAVE);
        int oi = __shfl_xor(besti, off, WAVE);
        if (ov > best || (ov == best && oi >= 0 && (besti < 0 || oi < besti))) {
            best = ov;
            besti = oi;
        }
    }
    __shared__ float sval[4];
    __shared__ int sidx[4];
    const int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) {
        sval[wid] = best;
        sidx[wid] = besti;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        for (int wv = 1; wv < (int)(blockDim.x / WAVE); ++wv) {
            if (sval[wv] > best || (sval[wv] == best && sidx[wv] >= 0 &&
                                    (besti < 0 || sidx[wv] < besti))) {
                best = sval[wv];
                besti = sidx[wv];
            }
        }
        out[b] = besti;
    }
}

extern "C" int oa_masked_argmax(void* stream, const void* logits, const void* mask,
                                void* out, int B, int V) {
    if (V % 8 != 0) return -100;
    const int mask_words = CEIL_DIV(V, 32);
    hipLaunchKernelGGL(masked_argmax_kernel, dim3(B), dim3(256), 0,
                       (hipStream_t)stream, (const uint32_t*)logits,
                       (const uint32_t*)mask, (int*)out, V, mask_words);
    HIP_CHECK_LAUNCH();
    return 0;
}
