"""Single-shape TN wgrad loop for PMC / A-B runs.
Usage: [PDT_TN_TILE=128|256] python tools/tn_bench.py [I J iters]
"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_ddp_template_amd.ops.native import native
EXT = native()
M = 256 * 197
I = int(sys.argv[1]) if len(sys.argv) > 1 else 3072
J = int(sys.argv[2]) if len(sys.argv) > 2 else 768
iters = int(sys.argv[3]) if len(sys.argv) > 3 else 20
dy = torch.randn(M, I, device="cuda", dtype=torch.bfloat16)
x = torch.randn(M, J, device="cuda", dtype=torch.bfloat16)
for _ in range(3):
    EXT.gemm_tn(dy, x)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(iters):
    EXT.gemm_tn(dy, x)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
tf = 2.0 * M * I * J / dt / 1e12
print(f"TILE={os.environ.get('PDT_TN_TILE','def')} I={I} J={J}: "
      f"{dt*1e3:.3f} ms  {tf:.0f} TF")
