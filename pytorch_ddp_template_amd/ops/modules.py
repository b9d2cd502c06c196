"""nn.Module wrappers over the HIP-backed functional ops.

These mirror the torch.nn API surface (parameter names, state_dict layout)
so checkpoints look conventional, but dispatch to the CDNA4 kernels.
Conv/BN modules are NHWC (see ops/functional.py).
"""

from __future__ import annotations

import math

import torch
from torch import nn

from . import functional as X


class Linear(nn.Module):
    def __init__(self, in_features, out_features, bias=True, act=None):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.act = act
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.empty(out_features)) if bias else None
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if self.bias is not None:
            bound = 1.0 / math.sqrt(self.in_features)
            nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x):
        return X.linear(x, self.weight, self.bias, self.act)

    def extra_repr(self):
        return (
            f"in_features={self.in_features}, out_features={self.out_features},"
            f" bias={self.bias is not None}, act={self.act}"
        )


class ReLU(nn.Module):
    def forward(self, x):
        return X.relu(x)


class GELU(nn.Module):
    def forward(self, x):
        return X.gelu(x)


class Conv2dNHWC(nn.Module):
    """Conv2d on (N, H, W, C) tensors; weight stored (K, R, S, C)."""

    def __init__(
        self, in_ch, out_ch, kernel_size, stride=1, padding=0, bias=False, act=None
    ):
        super().__init__()
        k = (
            (kernel_size, kernel_size)
            if isinstance(kernel_size, int)
            else tuple(kernel_size)
        )
        self.in_ch, self.out_ch = in_ch, out_ch
        self.kernel_size, self.stride, self.padding = k, stride, padding
        self.act = act
        self.weight = nn.Parameter(torch.empty(out_ch, k[0], k[1], in_ch))
        self.bias = nn.Parameter(torch.empty(out_ch)) if bias else None
        self.reset_parameters()

    def reset_parameters(self):
        fan_in = self.in_ch * self.kernel_size[0] * self.kernel_size[1]
        nn.init.normal_(self.weight, 0.0, math.sqrt(2.0 / fan_in))
        if self.bias is not None:
            nn.init.zeros_(self.bias)

    def forward(self, x):
        return X.conv2d_nhwc(
            x, self.weight, self.bias, self.stride, self.padding, self.act
        )

    def extra_repr(self):
        return (
            f"{self.in_ch}, {self.out_ch}, kernel_size={self.kernel_size},"
            f" stride={self.stride}, padding={self.padding}, act={self.act}"
        )


class BatchNorm2dNHWC(nn.Module):
    def __init__(self, num_features, eps=1e-5, momentum=0.1, act=None):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.act = act
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        # tracked count lives host-side (a GPU scalar += here cost a kernel
        # launch per BN layer per step); it enters state_dict via the
        # _save/_load hooks below so checkpoints still carry it.
        self._batches_tracked = 0

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        super()._save_to_state_dict(destination, prefix, keep_vars)
        destination[prefix + "num_batches_tracked"] = torch.tensor(
            self._batches_tracked, dtype=torch.long
        )

    def _load_from_state_dict(self, state_dict, prefix, *args, **kwargs):
        key = prefix + "num_batches_tracked"
        if key in state_dict:
            self._batches_tracked = int(state_dict.pop(key))
        super()._load_from_state_dict(state_dict, prefix, *args, **kwargs)

    def _apply(self, fn, recurse=True):
        # Running stats stay fp32 regardless of model dtype: the BN kernels
        # accumulate and update them in fp32 (model.to(bf16) must not
        # downcast them).
        super()._apply(fn, recurse)
        self.running_mean.data = self.running_mean.data.float()
        self.running_var.data = self.running_var.data.float()
        return self

    def forward(self, x):
        if self.training:
            self._batches_tracked += 1
        return X.batch_norm2d_nhwc(
            x,
            self.weight,
            self.bias,
            self.running_mean,
            self.running_var,
            self.training,
            self.momentum,
            self.eps,
            self.act,
        )

    def extra_repr(self):
        return f"{self.num_features}, eps={self.eps}, momentum={self.momentum}, act={self.act}"


class LayerNorm(nn.Module):
    def __init__(self, dim, eps=1e-6):
        super().__init__()
        self.dim = dim
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))

    def forward(self, x):
        return X.layer_norm(x, self.weight, self.bias, self.eps)


class GlobalAvgPoolNHWC(nn.Module):
    def forward(self, x):
        return X.global_avg_pool_nhwc(x)


class MaxPool2dNHWC(nn.Module):
    def __init__(self, kernel_size, stride, padding=0):
        super().__init__()
        self.kernel_size, self.stride, self.padding = kernel_size, stride, padding

    def forward(self, x):
        return X.max_pool2d_nhwc(x, self.kernel_size, self.stride, self.padding)


class MSELoss(nn.Module):
    """The reference's criterion (ddp.py:164)."""

    def forward(self, pred, target):
        return X.mse_loss(pred, target)


class CrossEntropyLoss(nn.Module):
    """Fused log-softmax + NLL, mean reduction."""

    def forward(self, logits, target):
        return X.cross_entropy(logits, target)
