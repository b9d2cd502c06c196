"""Per-shape wgrad timing on the r18/CIFAR conv set (batch 8192)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_ddp_template_amd.ops.native import native
EXT = native(); DEV = "cuda:0"

def bench(fn, iters=10, warmup=3):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3

B = 8192
shapes = [  # (name, h_in, c, k, r, stride, pad, count_per_step)
    ("L1 3x3", 32, 64, 64, 3, 1, 1, 4),
    ("L2 3x3s2", 32, 64, 128, 3, 2, 1, 1),
    ("L2 3x3", 16, 128, 128, 3, 1, 1, 3),
    ("L2 1x1s2", 32, 64, 128, 1, 2, 0, 1),
    ("L3 3x3s2", 16, 128, 256, 3, 2, 1, 1),
    ("L3 3x3", 8, 256, 256, 3, 1, 1, 3),
    ("L3 1x1s2", 16, 128, 256, 1, 2, 0, 1),
    ("L4 3x3s2", 8, 256, 512, 3, 2, 1, 1),
    ("L4 3x3", 4, 512, 512, 3, 1, 1, 3),
    ("L4 1x1s2", 8, 256, 512, 1, 2, 0, 1),
]
tot = 0.0
for name, h, c, k, r, s, p, cnt in shapes:
    ho = (h + 2 * p - r) // s + 1
    dy = (torch.randn(B, ho, ho, k, device=DEV) * 0.1).to(torch.bfloat16)
    x = torch.randn(B, h, h, c, device=DEV).to(torch.bfloat16)
    t = bench(lambda: EXT.conv2d_wgrad(dy, x, s, p, r, r))
    fl = 2.0 * B * ho * ho * k * r * r * c
    gb = (B * ho * ho * k + B * h * h * c) * 2 / 1e9
    tot += t * cnt
    print(f"{name:9s} {t:7.3f} ms x{cnt}  {fl/t/1e9:6.0f} TF  io {gb/t*1e3:5.0f} GB/s", flush=True)
    del dy, x
print(f"total wgrad/step ~ {tot:.1f} ms")
