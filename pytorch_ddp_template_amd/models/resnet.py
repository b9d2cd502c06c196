"""ResNet-18/50 in NHWC on the framework's HIP-backed ops.

Built for the BASELINE.json benchmark configs (ResNet-18 on CIFAR-shape
3x32x32, ResNet-50 on ImageNet-shape 3x224x224).  Architecture follows the
standard torchvision ResNet v1 definitions; layout is NHWC end-to-end and
conv->bn->relu chains use the fused BN+ReLU epilogue; in the fused-block
training path the residual add + final ReLU ride in the last BN's normalize
epilogue (ops/fused_block.py).  The reference template has no conv models at
all (FooModel is a 2-layer MLP, reference model.py:8-16) —
these exist for the BASELINE.json benchmark configs.

Two stems:
* ``stem="cifar"`` — 3x3 s1 conv, no maxpool (the standard CIFAR ResNet stem;
  a 7x7 s2 stem on 32x32 inputs would collapse the spatial dims).
* ``stem="imagenet"`` — 7x7 s2 conv + 3x3 s2 maxpool.
"""

from __future__ import annotations

import torch
from torch import nn

from ..ops import (
    BatchNorm2dNHWC,
    Conv2dNHWC,
    GlobalAvgPoolNHWC,
    Linear,
    MaxPool2dNHWC,
    add_relu,
)
from ..ops.fused_block import fused_basic_block, fused_bottleneck
from ..ops.native import native_available


def _conv_bn(in_ch, out_ch, k, stride, pad, act=None):
    return nn.Sequential(
        Conv2dNHWC(in_ch, out_ch, k, stride=stride, padding=pad, bias=False),
        BatchNorm2dNHWC(out_ch, act=act),
    )


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_ch, ch, stride=1):
        super().__init__()
        self.stride = stride
        self.conv1 = _conv_bn(in_ch, ch, 3, stride, 1, act="relu")
        self.conv2 = _conv_bn(ch, ch, 3, 1, 1)
        self.down = (
            _conv_bn(in_ch, ch * self.expansion, 1, stride, 0)
            if (stride != 1 or in_ch != ch * self.expansion)
            else None
        )

    def _can_fuse(self, x):
        # single-autograd-node fused path (ops/fused_block.py): training-mode
        # GPU bf16/f16 with pow2 channels (the fast conv kernels' domain)
        if not (self.training and x.is_cuda and torch.is_grad_enabled()):
            return False
        if x.dtype not in (torch.bfloat16, torch.float16):
            return False
        c_in, c_out = x.shape[-1], self.conv1[0].weight.shape[0]
        pow2 = lambda v: v >= 8 and (v & (v - 1)) == 0  # noqa: E731
        return pow2(c_in) and pow2(c_out) and native_available()

    def forward(self, x):
        if self._can_fuse(x):
            for m in (self.conv1[1], self.conv2[1]) + (
                (self.down[1],) if self.down is not None else ()
            ):
                m._batches_tracked += 1
            return fused_basic_block(
                x, self.conv1[0], self.conv1[1], self.conv2[0], self.conv2[1],
                self.down[0] if self.down is not None else None,
                self.down[1] if self.down is not None else None,
                self.stride,
            )
        idt = x if self.down is None else self.down(x)
        out = self.conv2(self.conv1(x))
        return add_relu(out, idt)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, ch, stride=1):
        super().__init__()
        self.stride = stride
        self.conv1 = _conv_bn(in_ch, ch, 1, 1, 0, act="relu")
        self.conv2 = _conv_bn(ch, ch, 3, stride, 1, act="relu")
        self.conv3 = _conv_bn(ch, ch * self.expansion, 1, 1, 0)
        self.down = (
            _conv_bn(in_ch, ch * self.expansion, 1, stride, 0)
            if (stride != 1 or in_ch != ch * self.expansion)
            else None
        )

    _can_fuse = BasicBlock._can_fuse

    def forward(self, x):
        if self._can_fuse(x):
            bns = [self.conv1[1], self.conv2[1], self.conv3[1]] + (
                [self.down[1]] if self.down is not None else []
            )
            for m in bns:
                m._batches_tracked += 1
            return fused_bottleneck(
                x, self.conv1[0], self.conv1[1], self.conv2[0], self.conv2[1],
                self.conv3[0], self.conv3[1],
                self.down[0] if self.down is not None else None,
                self.down[1] if self.down is not None else None,
                self.stride,
            )
        idt = x if self.down is None else self.down(x)
        out = self.conv3(self.conv2(self.conv1(x)))
        return add_relu(out, idt)


class ResNet(nn.Module):
    def __init__(self, block, layers, num_classes=10, stem="cifar"):
        super().__init__()
        self.stem_kind = stem
        if stem == "cifar":
            self.stem = nn.Sequential(_conv_bn(3, 64, 3, 1, 1, act="relu"))
        elif stem == "imagenet":
            self.stem = nn.Sequential(
                _conv_bn(3, 64, 7, 2, 3, act="relu"),
                MaxPool2dNHWC(3, 2, 1),
            )
        else:
            raise ValueError(f"unknown stem {stem!r}")
        self.in_ch = 64
        self.layer1 = self._make_layer(block, 64, layers[0], 1)
        self.layer2 = self._make_layer(block, 128, layers[1], 2)
        self.layer3 = self._make_layer(block, 256, layers[2], 2)
        self.layer4 = self._make_layer(block, 512, layers[3], 2)
        self.pool = GlobalAvgPoolNHWC()
        self.fc = Linear(512 * block.expansion, num_classes)

    def _make_layer(self, block, ch, n, stride):
        blocks = [block(self.in_ch, ch, stride)]
        self.in_ch = ch * block.expansion
        for _ in range(1, n):
            blocks.append(block(self.in_ch, ch, 1))
        return nn.Sequential(*blocks)

    def forward(self, x):
        # x: (N, H, W, 3) NHWC
        if x.dim() == 4 and x.shape[1] == 3 and x.shape[-1] != 3:
            # accept NCHW input for convenience; convert once at entry
            x = x.permute(0, 2, 3, 1).contiguous()
        x = self.stem(x)
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.pool(x)
        return self.fc(x)


def resnet18(num_classes=10, stem="cifar"):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, stem)


def resnet50(num_classes=1000, stem="imagenet"):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, stem)
