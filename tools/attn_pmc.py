import sys, time
sys.path.insert(0, '/root/repo')
import torch
from pytorch_ddp_template_amd.ops.native import native
EXT = native()
N, H, S, dh = 256, 12, 197, 64
qkv = torch.randn(N, S, 3 * H * dh, device='cuda', dtype=torch.bfloat16)
for _ in range(3):
    out, P, stats = EXT.attn_fwd(qkv, H, 0.125, True)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    out, P, stats = EXT.attn_fwd(qkv, H, 0.125, True)
torch.cuda.synchronize()
print('attn_fwd ms:', (time.perf_counter() - t0) / 10 * 1e3)
