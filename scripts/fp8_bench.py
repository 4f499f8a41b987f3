import argparse, time, torch, sys
import os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from opsagent_amd import ops
from opsagent_amd.ops import hip_lib
lib = hip_lib.get_lib()
def t(fn, iters=20, warm=5):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters
ap = argparse.ArgumentParser(); ap.add_argument("--shapes", default=""); args = ap.parse_args()
SH = [(1024,8192,8192),(1024,10240,8192),(1024,57344,8192),(1024,8192,28672),(2048,4096,4096)]
if args.shapes: SH = [SH[int(i)] for i in args.shapes.split(",")]
for (M,N,K) in SH:
    x = torch.randn(M,K,dtype=torch.bfloat16,device="cuda")*0.3
    w = torch.randn(N,K,dtype=torch.bfloat16,device="cuda")*0.3
    wq,ws = ops.quant_fp8(w)
    xq,xs = ops.quant_fp8(x)
    out = torch.empty(M,N,dtype=torch.bfloat16,device="cuda")
    def g():
        rc = lib.oa_gemm_fp8(hip_lib.current_stream_ptr(), xq.data_ptr(), wq.data_ptr(), xs.data_ptr(), ws.data_ptr(), out.data_ptr(), M,N,K)
        assert rc==0
    tf = 2*M*N*K
    tg = t(g); tb = t(lambda: torch.nn.functional.linear(x,w))
    # correctness vs dequant ref on same operands
    g(); torch.cuda.synchronize()
    import opsagent_amd.ops.torch_ref as tr
    xd = tr.dequant_fp8(xq[:64].cpu(), xs[:64].cpu()); wd = tr.dequant_fp8(wq.cpu(), ws.cpu())
    ref = (xd @ wd.T)
    err = (out[:64].float().cpu() - ref).abs().max().item()
    print(f"M{M} N{N} K{K}: fp8GEMM {tf/1e12/tg:.0f} TF {tg*1e3:.2f}ms | blaslt bf16 {tf/1e12/tb:.0f} TF {tb*1e3:.2f}ms | err {err:.3f}")
