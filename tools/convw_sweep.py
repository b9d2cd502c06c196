"""conv wgrad block-target sweep (r18 shapes at bench-like batch)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_ddp_template_amd.ops.native import native
EXT = native()
N = int(os.environ.get("SWEEP_N", "2048"))
for C, K, H in ((64, 64, 32), (128, 128, 16), (256, 256, 8), (512, 512, 4)):
    x = torch.randn(N, H, H, C, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(N, H, H, K, device="cuda", dtype=torch.bfloat16)
    for _ in range(3):
        EXT.conv2d_wgrad(dy, x, 1, 1, 3, 3)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    it = 10
    for _ in range(it):
        EXT.conv2d_wgrad(dy, x, 1, 1, 3, 3)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / it
    tf = 2.0 * N * H * H * K * C * 9 / dt / 1e12
    print(f"B={os.environ.get('PDT_CONV_WGRAD_B','512')} C{C} H{H}: "
          f"{dt*1e3:7.3f} ms {tf:5.0f} TF", flush=True)
