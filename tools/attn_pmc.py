"""attn fwd/bwd microbench (timing + PMC target).

python tools/attn_pmc.py          — wall timings per call
rocprofv3 --pmc ... -- python tools/attn_pmc.py pmc   — counter run (few iters)
"""
import sys, time
sys.path.insert(0, '/root/repo')
import torch
from pytorch_ddp_template_amd.ops.native import native
EXT = native()
N, H, S, dh = 256, 12, 197, 64
pmc = len(sys.argv) > 1 and sys.argv[1] == "pmc"
iters = 2 if pmc else 10
qkv = torch.randn(N, S, 3 * H * dh, device='cuda', dtype=torch.bfloat16)
dout = torch.randn(N, S, H * dh, device='cuda', dtype=torch.bfloat16)


def bench(fn, label):
    for _ in range(2 if pmc else 3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print(f'{label}: {(time.perf_counter() - t0) / iters * 1e3:.3f} ms')


out, _, stats = EXT.attn_fwd(qkv, H, 0.125, False)
bench(lambda: EXT.attn_fwd(qkv, H, 0.125, False), 'attn_fwd (no P)')
if not pmc:
    bench(lambda: EXT.attn_fwd(qkv, H, 0.125, True), 'attn_fwd (want P)')
bench(lambda: EXT.attn_bwd(qkv, dout, out, stats, H, 0.125), 'attn_bwd')
