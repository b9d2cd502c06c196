"""FooModel — the reference's 2-layer MLP (reference model.py:8-16).

Linear(10,10) -> ReLU -> Linear(10,5), rebuilt on the framework's
HIP-backed Linear (MFMA GEMM on gfx950; fused ReLU epilogue on the
first layer instead of a separate elementwise pass).
"""

from __future__ import annotations

from torch import nn

from ..ops import Linear


class FooModel(nn.Module):
    def __init__(self, in_features: int = 10, hidden: int = 10, out_features: int = 5):
        super().__init__()
        # Fused GEMM+bias+ReLU replaces the reference's separate
        # net1 -> relu (model.py:11-12); net2 is a plain GEMM+bias.
        self.net1 = Linear(in_features, hidden, act="relu")
        self.relu = nn.Identity()  # kept for state-dict/shape parity with model.py:12
        self.net2 = Linear(hidden, out_features)

    def forward(self, x):
        return self.net2(self.net1(x))
