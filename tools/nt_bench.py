"""Single-shape NT (linear fwd) loop for PMC / A-B runs.
Usage: python tools/nt_bench.py [N K iters]  (M fixed = 50432)
"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_ddp_template_amd.ops.native import native
EXT = native()
M = 256 * 197
N = int(sys.argv[1]) if len(sys.argv) > 1 else 3072
K = int(sys.argv[2]) if len(sys.argv) > 2 else 768
iters = int(sys.argv[3]) if len(sys.argv) > 3 else 20
x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
for _ in range(3):
    EXT.gemm_nt(x, w, None, False, False)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(iters):
    EXT.gemm_nt(x, w, None, False, False)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
print(f"NT M={M} N={N} K={K}: {dt*1e3:.3f} ms  {2.0*M*N*K/dt/1e12:.0f} TF")
