"""ResNet-18 (bf16) — the headline FedAvg benchmark model.

Hand-written (no torchvision in the image): standard BasicBlock ResNet-18
for 224×224 inputs. The residual 3×3 convs and 1×1 downsamples run on the
hand-written NHWC implicit-GEMM kernel (ops/conv.py, forward; backward is
MIOpen for now), the 3-channel 7×7 stem stays on MIOpen; the classifier
head runs on the hand-written MFMA GEMM (ops.gemm.Linear); BatchNorm is
the fused NHWC kernel set (ops/batchnorm.py).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.conv import Conv2d as HipConv2d
from ..ops.conv import MaxPool2d as HipMaxPool2d

from ..ops.gemm import Linear as MfmaLinear
from ..ops.batchnorm import FusedBatchNorm2d


class BatchNorm2dBF16(nn.Module):
    """BatchNorm with bf16 affine params (flat-param-space compatible) and
    fp32 statistics/compute (stability). Forward casts around F.batch_norm."""

    def __init__(self, num_features: int, eps: float = 1e-5, momentum: float = 0.1):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(num_features, dtype=torch.bfloat16))
        self.bias = nn.Parameter(torch.zeros(num_features, dtype=torch.bfloat16))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.eps, self.momentum = eps, momentum

    def forward(self, x):
        # bf16 activations stay bf16 (native_batch_norm takes fp32 affine
        # params with bf16 input — autocast's convention); only the tiny
        # per-channel params are cast. Avoids 2 full fp32 copies of x.
        return torch.nn.functional.batch_norm(
            x, self.running_mean, self.running_var,
            self.weight.float(), self.bias.float(),
            self.training, self.momentum, self.eps,
        )


def conv3x3(cin, cout, stride=1):
    # hand-written NHWC implicit-GEMM kernel on GPU (ops/conv.py)
    return HipConv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.conv1 = conv3x3(cin, cout, stride)
        self.bn1 = FusedBatchNorm2d(cout)
        self.conv2 = conv3x3(cout, cout)
        self.bn2 = FusedBatchNorm2d(cout)
        # fuse each BN's batch-stat accumulation into the producing conv's
        # epilogue (skips BN-fwd's whole partials read pass); plain
        # attribute set so the BN is not re-registered as a conv child
        object.__setattr__(self.conv1, "stats_bn", self.bn1)
        object.__setattr__(self.conv2, "stats_bn", self.bn2)
        self.down_conv = None
        if stride != 1 or cin != cout:
            self.down_conv = HipConv2d(cin, cout, 1, stride=stride, bias=False)
            self.down_bn = FusedBatchNorm2d(cout)
            object.__setattr__(self.down_conv, "stats_bn", self.down_bn)

    def forward(self, x):
        if self.down_conv is None:
            idt = x
        else:
            idt = self.down_bn(self.down_conv(x))
        out = self.bn1(self.conv1(x), relu=True)
        # bn2 + residual add + relu fused into one pass
        return self.bn2(self.conv2(out), residual=idt, relu=True)


class Bottleneck(nn.Module):
    """ResNet-50-style 1x1 -> 3x3 -> 1x1 block, conv-epilogue BN-stat
    fusion on every conv like BasicBlock (all three shapes run on the
    hand-written NHWC conv kernels: 1x1 and 3x3 dispatch paths)."""

    expansion = 4

    def __init__(self, cin, width, stride=1):
        super().__init__()
        cout = width * self.expansion
        self.conv1 = HipConv2d(cin, width, 1, bias=False)
        self.bn1 = FusedBatchNorm2d(width)
        self.conv2 = conv3x3(width, width, stride)
        self.bn2 = FusedBatchNorm2d(width)
        self.conv3 = HipConv2d(width, cout, 1, bias=False)
        self.bn3 = FusedBatchNorm2d(cout)
        for c, b in ((self.conv1, self.bn1), (self.conv2, self.bn2),
                     (self.conv3, self.bn3)):
            object.__setattr__(c, "stats_bn", b)
        self.down_conv = None
        if stride != 1 or cin != cout:
            self.down_conv = HipConv2d(cin, cout, 1, stride=stride, bias=False)
            self.down_bn = FusedBatchNorm2d(cout)
            object.__setattr__(self.down_conv, "stats_bn", self.down_bn)

    def forward(self, x):
        idt = x if self.down_conv is None else self.down_bn(self.down_conv(x))
        out = self.bn1(self.conv1(x), relu=True)
        out = self.bn2(self.conv2(out), relu=True)
        return self.bn3(self.conv3(out), residual=idt, relu=True)


class ResNet18(nn.Module):
    def __init__(self, num_classes: int = 1000, use_mfma_head: bool = True):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = FusedBatchNorm2d(64)
        self.maxpool = HipMaxPool2d()  # hand-written NHWC kernels
        layers = []
        cin = 64
        for cout, stride in ((64, 1), (128, 2), (256, 2), (512, 2)):
            layers.append(BasicBlock(cin, cout, stride))
            layers.append(BasicBlock(cout, cout))
            cin = cout
        self.layers = nn.Sequential(*layers)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        if use_mfma_head:
            self.fc = MfmaLinear(512, num_classes)
        else:
            self.fc = nn.Linear(512, num_classes)

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x), relu=True))
        x = self.layers(x)
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


class ResNet50(nn.Module):
    def __init__(self, num_classes: int = 1000, use_mfma_head: bool = True):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = FusedBatchNorm2d(64)
        self.maxpool = HipMaxPool2d()
        layers = []
        cin = 64
        for width, blocks, stride in ((64, 3, 1), (128, 4, 2),
                                      (256, 6, 2), (512, 3, 2)):
            layers.append(Bottleneck(cin, width, stride))
            cin = width * Bottleneck.expansion
            for _ in range(blocks - 1):
                layers.append(Bottleneck(cin, width))
        self.layers = nn.Sequential(*layers)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = (MfmaLinear(2048, num_classes) if use_mfma_head
                   else nn.Linear(2048, num_classes))

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x), relu=True))
        x = self.layers(x)
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet50(num_classes: int = 1000, dtype=torch.bfloat16, device="cpu",
             use_mfma_head: bool | None = None) -> ResNet50:
    if use_mfma_head is None:
        use_mfma_head = True
    m = ResNet50(num_classes, use_mfma_head=use_mfma_head)
    m = m.to(device=device, dtype=dtype)
    for mod in m.modules():
        if isinstance(mod, (BatchNorm2dBF16, FusedBatchNorm2d)):
            mod.running_mean.data = mod.running_mean.data.float()
            mod.running_var.data = mod.running_var.data.float()
    return m


def resnet18(num_classes: int = 1000, dtype=torch.bfloat16, device="cpu",
             use_mfma_head: bool | None = None) -> ResNet18:
    if use_mfma_head is None:
        # MFMA head requires CUDA(HIP) extension or CPU reference — both exist
        use_mfma_head = True
    m = ResNet18(num_classes, use_mfma_head=use_mfma_head)
    m = m.to(device=device, dtype=dtype)
    # keep BN statistics buffers fp32 regardless of model dtype
    for mod in m.modules():
        if isinstance(mod, (BatchNorm2dBF16, FusedBatchNorm2d)):
            mod.running_mean.data = mod.running_mean.data.float()
            mod.running_var.data = mod.running_var.data.float()
    return m
