import torch, sys
sys.path.insert(0, "/root/repo")
from opsagent_amd import ops
from opsagent_amd.ops import torch_ref, hip_lib

dev = "cuda"
lib = hip_lib.get_lib()
M = N = 128; K = 512
# A = identity in top-left 64x64, B asymmetric
A = torch.zeros(M, K, dtype=torch.bfloat16, device=dev)
for i in range(128): A[i, i] = 1.0
B = torch.zeros(N, K, dtype=torch.bfloat16, device=dev)
for n in range(N):
    for k in range(K):
        B[n, k] = ((n * 7 + k * 3) % 13 - 6) / 4.0
aq, asc = ops.quant_fp8(A.contiguous())
bq, bsc = ops.quant_fp8(B.contiguous())
C = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
rc = lib.oa_gemm_fp8(hip_lib.current_stream_ptr(), aq.data_ptr(), bq.data_ptr(),
                     asc.data_ptr(), bsc.data_ptr(), C.data_ptr(), M, N, K)
torch.cuda.synchronize()
ref = (torch_ref.dequant_fp8(aq.cpu(), asc.cpu()) @ torch_ref.dequant_fp8(bq.cpu(), bsc.cpu()).T)
got = C.float().cpu()
print("rc", rc)
print("ref[0,:8] ", ref[0,:8].tolist())
print("got[0,:8] ", got[0,:8].tolist())
print("ref[:8,0] ", ref[:8,0].tolist())
print("got[:8,0] ", got[:8,0].tolist())
print("ref[5,:8] ", ref[5,:8].tolist())
print("got[5,:8] ", got[5,:8].tolist())
# where does ref[0,0] appear in got?
tgt = ref[0,3].item()
loc = (got - tgt).abs() < 1e-3
print("ref[0,3] found at", loc.nonzero()[:5].tolist())
err = (got - ref).abs()
print("max err", err.max().item(), "frac wrong", (err > 0.05).float().mean().item())
