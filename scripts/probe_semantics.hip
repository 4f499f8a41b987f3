// One-off semantics probe for gfx950 idioms used by attn_prefill_v3:
//   * v_permlane32_swap_b32 half-exchange direction
//   * v_cvt_pk_bf16_f32 operand->lo/hi order
// Build + run on a GPU box:
//   hipcc --offload-arch=gfx950 -O2 scripts/probe_semantics.hip -o /tmp/sem && /tmp/sem
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void probe(unsigned* perm_a, unsigned* perm_b, unsigned* pk) {
    int lane = threadIdx.x;
    unsigned v = 1000 + lane;
    unsigned a = v, b = v;
    asm volatile("v_permlane32_swap_b32 %0, %1" : "+v"(a), "+v"(b));
    perm_a[lane] = a;
    perm_b[lane] = b;
    float f0 = 1.5f, f1 = -2.25f;
    unsigned r;
    asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(f0), "v"(f1));
    pk[lane] = r;
}

int main() {
    unsigned *pa, *pb, *pkd;
    hipMalloc(&pa, 64 * 4);
    hipMalloc(&pb, 64 * 4);
    hipMalloc(&pkd, 64 * 4);
    probe<<<1, 64>>>(pa, pb, pkd);
    unsigned ha[64], hb[64], hk[64];
    hipMemcpy(ha, pa, 64 * 4, hipMemcpyDeviceToHost);
    hipMemcpy(hb, pb, 64 * 4, hipMemcpyDeviceToHost);
    hipMemcpy(hk, pkd, 64 * 4, hipMemcpyDeviceToHost);
    printf("lane0:  a=%u b=%u\nlane32: a=%u b=%u\nlane5:  a=%u b=%u\nlane37: a=%u b=%u\n",
           ha[0], hb[0], ha[32], hb[32], ha[5], hb[5], ha[37], hb[37]);
    printf("pk = 0x%08x (lo=0x%04x hi=0x%04x; 1.5=0x3fc0, -2.25=0xc010)\n",
           hk[0], hk[0] & 0xffff, hk[0] >> 16);
    return 0;
}
