"""ViT-B/16 on the framework's HIP-backed ops (BASELINE.json config 5).

Attention is composed from the framework's batched MFMA GEMM primitives
(``bmm_nt`` / ``bmm_nn``) plus a fused row-softmax kernel — every
matmul-shaped op runs on hand-written CDNA4 kernels, with the score matrix
materialized (S=197 is small; a fused flash-style kernel is a later
optimization, see SURVEY.md §7 rung 5).

Patch embedding is a Linear over unfolded 16x16x3 patches (= 768-dim
vectors), which on NHWC input is a pure reshape + one GEMM — no conv needed.
"""

from __future__ import annotations

import math

import torch
from torch import nn

from ..ops import GELU, LayerNorm, Linear
from ..ops.functional import attention, attention_qkv
from ..ops.native import native_available


class MultiHeadAttention(nn.Module):
    def __init__(self, dim, heads):
        super().__init__()
        assert dim % heads == 0
        self.dim, self.heads = dim, heads
        self.qkv = Linear(dim, 3 * dim)
        self.proj = Linear(dim, dim)

    def forward(self, x):
        # x: (N, S, D)
        N, S, D = x.shape
        h, dh = self.heads, D // self.heads
        qkv = self.qkv(x)  # (N, S, 3D)
        # the flash forward/backward tile K/V, so S is uncapped (the old
        # S<=224 limit belonged to the P-materializing two-pass kernel;
        # fused-vs-composed grads are tested at S=320)
        if (
            x.is_cuda and dh == 64
            and x.dtype in (torch.bfloat16, torch.float16)
            and native_available()
        ):
            # fused kernel: stages q/k/v straight from the packed qkv
            out = attention_qkv(qkv.contiguous(), h, 1.0 / math.sqrt(dh))
            return self.proj(out)
        qkv = qkv.reshape(N, S, 3, h, dh).permute(2, 0, 3, 1, 4)
        q, k, v = (t.reshape(N * h, S, dh).contiguous() for t in qkv)
        out = attention(q, k, v, 1.0 / math.sqrt(dh))  # (N*h, S, dh)
        out = out.reshape(N, h, S, dh).permute(0, 2, 1, 3).reshape(N, S, D)
        return self.proj(out)


class Block(nn.Module):
    def __init__(self, dim, heads, mlp_ratio=4):
        super().__init__()
        self.norm1 = LayerNorm(dim)
        self.attn = MultiHeadAttention(dim, heads)
        self.norm2 = LayerNorm(dim)
        self.mlp = nn.Sequential(
            Linear(dim, dim * mlp_ratio),
            GELU(),
            Linear(dim * mlp_ratio, dim),
        )

    def forward(self, x):
        x = x + self.attn(self.norm1(x))
        x = x + self.mlp(self.norm2(x))
        return x


class ViT(nn.Module):
    def __init__(
        self,
        image_size=224,
        patch_size=16,
        dim=768,
        depth=12,
        heads=12,
        mlp_ratio=4,
        num_classes=1000,
        in_ch=3,
    ):
        super().__init__()
        assert image_size % patch_size == 0
        self.patch_size = patch_size
        self.grid = image_size // patch_size
        n_patches = self.grid * self.grid
        patch_dim = patch_size * patch_size * in_ch
        self.patch_embed = Linear(patch_dim, dim)
        self.cls_token = nn.Parameter(torch.zeros(1, 1, dim))
        self.pos_embed = nn.Parameter(torch.zeros(1, n_patches + 1, dim) )
        nn.init.trunc_normal_(self.pos_embed, std=0.02)
        nn.init.trunc_normal_(self.cls_token, std=0.02)
        self.blocks = nn.Sequential(*[Block(dim, heads, mlp_ratio) for _ in range(depth)])
        self.norm = LayerNorm(dim)
        self.head = Linear(dim, num_classes)

    def _patchify(self, x):
        # NHWC (N, H, W, C) -> (N, n_patches, P*P*C)
        N, H, W, C = x.shape
        P, G = self.patch_size, self.grid
        x = x.reshape(N, G, P, G, P, C).permute(0, 1, 3, 2, 4, 5)
        return x.reshape(N, G * G, P * P * C)

    def forward(self, x):
        if x.dim() == 4 and x.shape[1] == 3 and x.shape[-1] != 3:
            x = x.permute(0, 2, 3, 1).contiguous()
        x = self.patch_embed(self._patchify(x))
        cls = self.cls_token.to(x.dtype).expand(x.shape[0], -1, -1)
        x = torch.cat([cls, x], dim=1) + self.pos_embed.to(x.dtype)
        x = self.blocks(x)
        x = self.norm(x)
        return self.head(x[:, 0])


def vit_b16(num_classes=1000, image_size=224):
    return ViT(image_size=image_size, num_classes=num_classes)
