"""Kernel microbenchmark — isolates single hot ops for rocprofv3 runs.

Usage:  python tools/kbench.py [op ...]   (default: all)
Ops: wgrad1..wgrad4 (ResNet-18/CIFAR layer shapes), fwd1, dgrad2s, bnstats1.
Each op runs ITERS times on its layer shape so per-kernel PMC averages are
stable.  Keep ITERS small: PMC replay multiplies the cost.
"""

import sys

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from pytorch_ddp_template_amd.ops import native  # noqa: E402

ITERS = 10
N = 1024


def shapes(layer):
    # (C, K, H, stride) for resnet18-cifar blocks
    return {
        1: (64, 64, 32, 1),
        2: (128, 128, 16, 1),
        3: (256, 256, 8, 1),
        4: (512, 512, 4, 1),
    }[layer]


def run_wgrad(layer):
    C, K, H, _ = shapes(layer)
    x = torch.randn(N, H, H, C, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(N, H, H, K, device="cuda", dtype=torch.bfloat16)
    for _ in range(ITERS):
        native().conv2d_wgrad(dy, x, 1, 1, 3, 3)
    torch.cuda.synchronize()


def run_fwd(layer):
    C, K, H, _ = shapes(layer)
    x = torch.randn(N, H, H, C, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(K, 3, 3, C, device="cuda", dtype=torch.bfloat16)
    for _ in range(ITERS):
        native().conv2d_fwd(x, w, None, 1, 1, False)
    torch.cuda.synchronize()


def run_dgrad_stride2(layer):
    # stride-2 dgrad: layer2 first conv (64 -> 128, H 32 -> 16)
    Cp, K, H, _ = shapes(layer - 1)
    dy = torch.randn(N, H // 2, H // 2, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(K, 3, 3, Cp, device="cuda", dtype=torch.bfloat16)
    for _ in range(ITERS):
        native().conv2d_dgrad(dy, w, 2, 1, H, H)
    torch.cuda.synchronize()


def run_bnstats(layer):
    C, _, H, _ = shapes(layer)
    x = torch.randn(N * H * H, C, device="cuda", dtype=torch.bfloat16)
    g = torch.ones(C, device="cuda", dtype=torch.bfloat16)
    b = torch.zeros(C, device="cuda", dtype=torch.bfloat16)
    for _ in range(ITERS):
        native().bn_fwd(x, g, b, None, None, 0.1, 1e-5, True)
    torch.cuda.synchronize()


OPS = {
    "wgrad1": lambda: run_wgrad(1),
    "wgrad2": lambda: run_wgrad(2),
    "wgrad4": lambda: run_wgrad(4),
    "fwd1": lambda: run_fwd(1),
    "dgrad2s": lambda: run_dgrad_stride2(2),
    "bnstats1": lambda: run_bnstats(1),
}

if __name__ == "__main__":
    todo = sys.argv[1:] or list(OPS)
    for name in todo:
        OPS[name]()
        print(name, "done", flush=True)
