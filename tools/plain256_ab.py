"""Correctness + perf probe for the 256x256 glds plain-GEMM kernel."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_ddp_template_amd.ops.native import native

EXT = native()
DEV = "cuda:0"


def bench(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def check(m, k, n, bias=False, relu=False, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    a = torch.randn(m, k, generator=g).to(torch.bfloat16).to(DEV)
    b = torch.randn(n, k, generator=g).to(torch.bfloat16).to(DEV)
    bb = torch.randn(n, generator=g).to(torch.bfloat16).to(DEV) if bias else None
    out = EXT.gemm_nt(a, b, bb, relu, False)
    ref = a.float() @ b.float().t()
    if bias:
        ref = ref + bb.float()
    if relu:
        ref = torch.relu(ref)
    err = (out.float() - ref).abs().max().item()
    den = ref.abs().max().item()
    ok = err / max(den, 1.0) < 0.03
    print(f"check M{m} K{k} N{n} bias={bias} relu={relu}: "
          f"maxerr {err:.4g} / {den:.4g} {'OK' if ok else 'FAIL'}", flush=True)
    return ok


def perf(m, k, n, label=""):
    a = torch.randn(m, k, device=DEV).to(torch.bfloat16)
    b = torch.randn(n, k, device=DEV).to(torch.bfloat16)
    fl = 2.0 * m * k * n
    t_ours = bench(lambda: EXT.gemm_nt(a, b, None, False, False))
    t_blas = bench(lambda: a @ b.t())
    print(f"{label} M{m} K{k} N{n}: ours {t_ours:7.3f} ms "
          f"({fl/t_ours/1e9:6.0f} TF)  blas {t_blas:7.3f} ms "
          f"({fl/t_blas/1e9:6.0f} TF)", flush=True)


if __name__ == "__main__":
    ok = True
    ok &= check(4096, 768, 768)
    ok &= check(4096, 768, 768, bias=True)
    ok &= check(4096, 768, 768, bias=True, relu=True)
    ok &= check(4096, 4096, 4096, seed=1)
    ok &= check(50432, 768, 2304, seed=2)
    if not ok:
        sys.exit(1)
    perf(4096, 4096, 4096, "sq4k ")
    perf(8192, 8192, 8192, "sq8k ")
    M = 256 * 197
    # tail-free control shapes: grid = exact multiple of 256 CUs
    perf(65536, 768, 2304, "ntail")   # 256x9 = 2304 blocks = 9 waves exact
    perf(65536, 768, 768, "ntl2 ")    # 768 blocks = 3 waves exact
    perf(M, 768, 2304, "qkv  ")
    perf(M, 768, 768, "proj ")
    perf(M, 768, 3072, "mlp1 ")
    perf(M, 3072, 768, "mlp2 ")
