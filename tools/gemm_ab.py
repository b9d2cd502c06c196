"""A/B: framework NT/TN GEMM kernels vs rocBLAS (torch.matmul) on the
ViT-B/16 linear-layer shapes (batch 256, S=197 -> M=50432).

Run on a GPU box:  python tools/gemm_ab.py
"""
import sys, os, time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_ddp_template_amd.ops.native import native

EXT = native()
DEV = "cuda:0"


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def main():
    M = 256 * 197
    shapes = [
        ("qkv   768->2304", M, 768, 2304),
        ("proj  768->768 ", M, 768, 768),
        ("mlp1  768->3072", M, 768, 3072),
        ("mlp2  3072->768", M, 3072, 768),
    ]
    for name, m, k, n in shapes:
        x = torch.randn(m, k, device=DEV, dtype=torch.bfloat16)
        w = torch.randn(n, k, device=DEV, dtype=torch.bfloat16)
        dy = torch.randn(m, n, device=DEV, dtype=torch.bfloat16)
        bias = torch.randn(n, device=DEV, dtype=torch.bfloat16)
        flops_f = 2.0 * m * k * n

        t_ours_f = bench(lambda: EXT.gemm_nt(x, w, bias, False, False))
        t_blas_f = bench(lambda: torch.nn.functional.linear(x, w, bias))
        # dgrad: dx = dy @ w  (ours: explicit transpose + NT)
        t_ours_dx = bench(lambda: EXT.gemm_nt(dy, EXT.transpose2d(w), None, False, False))
        t_blas_dx = bench(lambda: dy @ w)
        # wgrad: dw = dy^T @ x
        t_ours_dw = bench(lambda: EXT.gemm_tn_bias(dy, x))
        t_blas_dw = bench(lambda: dy.t() @ x)

        def tf(t):
            return flops_f / (t * 1e-3) / 1e12

        print(f"{name}: fwd ours {t_ours_f:6.3f}ms ({tf(t_ours_f):5.0f}TF) "
              f"blas {t_blas_f:6.3f}ms ({tf(t_blas_f):5.0f}TF) | "
              f"dx ours {t_ours_dx:6.3f} blas {t_blas_dx:6.3f} | "
              f"dw ours {t_ours_dw:6.3f} blas {t_blas_dw:6.3f}")


if __name__ == "__main__":
    main()
