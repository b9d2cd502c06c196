"""A/B: TN wgrad kernels vs the transposed split-K NT route (ViT shapes)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_ddp_template_amd.ops.native import native
EXT = native(); DEV = "cuda:0"

def bench(fn, iters=20, warmup=5):
    for _ in range(warmup): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters * 1e3

M = 256 * 197
for name, n, k in [("qkv ", 2304, 768), ("proj", 768, 768),
                   ("mlp1", 3072, 768), ("mlp2", 768, 3072)]:
    dy = torch.randn(M, n, device=DEV).to(torch.bfloat16)
    x = torch.randn(M, k, device=DEV).to(torch.bfloat16)
    fl = 2.0 * M * n * k
    t_tn = bench(lambda: EXT.gemm_tn_bias(dy, x))
    def via_nt():
        dyT = EXT.transpose2d(dy)
        xT = EXT.transpose2d(x)
        return EXT.gemm_nt_splitk_f32(dyT, xT)
    t_nt = bench(via_nt)
    # correctness spot-check
    ref = (dy[:4096].float().t() @ x[:4096].float())
    got = EXT.gemm_nt_splitk_f32(EXT.transpose2d(dy[:4096].contiguous()),
                                 EXT.transpose2d(x[:4096].contiguous()))
    err = float((got - ref).abs().max() / ref.abs().max().clamp(min=1.0))
    print(f"{name} TN {t_tn:7.3f} ms ({fl/t_tn/1e9:5.0f} TF)  "
          f"NT-T {t_nt:7.3f} ms ({fl/t_nt/1e9:5.0f} TF)  err {err:.3g}",
          flush=True)
