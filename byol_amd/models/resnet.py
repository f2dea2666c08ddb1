"""ResNet encoder family, written from scratch for the MI355X BYOL engine.

Replicates the architecture the reference reaches through torchvision
(``/root/reference/main.py:190-193`` builds ``nn.Sequential(*children()[:-1])``
of a torchvision ResNet: the full network minus the final fc, keeping the
global average pool).  Built directly — no torchvision here — and shaped for
the MI355X hot path: every BatchNorm is a :class:`FusedBatchNorm` with the
following ReLU (and, at block exits, the residual add) folded into the same
HIP kernel, so a bottleneck block runs conv -> ONE fused bn+relu kernel pair
instead of conv -> BN stats -> BN norm -> ReLU -> add (see
``byol_amd/ops/bn.py``).  Numerics are identical to the torchvision module
tree (same stride/width/downsample placement, kaiming init, BN eps/momentum).

Supported archs: resnet18/34/50/101/152/200.
"""

from typing import List, Optional, Type, Union

import torch
import torch.nn as nn

from ..ops.bn import FusedBatchNorm

__all__ = ["ResNetEncoder", "build_encoder", "ARCH_SPECS", "arch_names"]


def conv3x3(in_planes: int, out_planes: int, stride: int = 1) -> nn.Conv2d:
    if in_planes % 32 == 0:
        # MFMA implicit-GEMM forward path (env-gated; MIOpen fallback)
        from ..ops.conv import MFMAConv3x3
        return MFMAConv3x3(in_planes, out_planes, kernel_size=3,
                           stride=stride, padding=1, bias=False)
    return nn.Conv2d(in_planes, out_planes, kernel_size=3, stride=stride,
                     padding=1, bias=False)


def conv1x1(in_planes: int, out_planes: int, stride: int = 1) -> nn.Conv2d:
    if stride == 1 and in_planes % 32 == 0:
        # hand-written f32 MFMA GEMM path with MIOpen fallback dispatch
        from ..ops.conv import MFMAConv1x1
        return MFMAConv1x1(in_planes, out_planes, kernel_size=1, bias=False)
    return nn.Conv2d(in_planes, out_planes, kernel_size=1, stride=stride,
                     bias=False)


class Downsample(nn.Module):
    """1x1 strided conv + (non-relu) BN on the identity branch."""

    def __init__(self, inplanes: int, outplanes: int, stride: int):
        super().__init__()
        self.conv = conv1x1(inplanes, outplanes, stride)
        self.bn = FusedBatchNorm(outplanes)

    def forward(self, x):
        return self.bn(self.conv(x))


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes: int, planes: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None):
        super().__init__()
        self.conv1 = conv3x3(inplanes, planes, stride)
        self.bn1 = FusedBatchNorm(planes, relu=True)
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = FusedBatchNorm(planes, relu=True)  # fused add+relu
        self.downsample = downsample
        self.stride = stride

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), residual=identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes: int, planes: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None, groups: int = 1,
                 base_width: int = 64):
        super().__init__()
        width = int(planes * (base_width / 64.0)) * groups
        self.conv1 = conv1x1(inplanes, width)
        self.bn1 = FusedBatchNorm(width, relu=True)
        if groups == 1:
            self.conv2 = conv3x3(width, width, stride)
        else:  # grouped 3x3 (ResNeXt): plain MIOpen path
            self.conv2 = nn.Conv2d(width, width, kernel_size=3,
                                   stride=stride, padding=1, groups=groups,
                                   bias=False)
        self.bn2 = FusedBatchNorm(width, relu=True)
        self.conv3 = conv1x1(width, planes * self.expansion)
        self.bn3 = FusedBatchNorm(planes * self.expansion,
                                  relu=True)  # fused add+relu
        self.downsample = downsample
        self.stride = stride

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=identity)


# arch -> (block, layers, groups, base_width); matches torchvision's specs
ARCH_SPECS = {
    "resnet18": (BasicBlock, [2, 2, 2, 2], 1, 64),
    "resnet34": (BasicBlock, [3, 4, 6, 3], 1, 64),
    "resnet50": (Bottleneck, [3, 4, 6, 3], 1, 64),
    "resnet101": (Bottleneck, [3, 4, 23, 3], 1, 64),
    "resnet152": (Bottleneck, [3, 8, 36, 3], 1, 64),
    "resnet200": (Bottleneck, [3, 24, 36, 3], 1, 64),
    "wide_resnet50_2": (Bottleneck, [3, 4, 6, 3], 1, 128),
    "wide_resnet101_2": (Bottleneck, [3, 4, 23, 3], 1, 128),
    "resnext50_32x4d": (Bottleneck, [3, 4, 6, 3], 32, 4),
    "resnext101_32x8d": (Bottleneck, [3, 4, 23, 3], 32, 8),
}


def arch_names() -> List[str]:
    return sorted(ARCH_SPECS.keys())


class ResNetEncoder(nn.Module):
    """ResNet minus the final fc: stem -> 4 stages -> global avgpool.

    Output is ``(B, out_channels, 1, 1)``; callers ``.view(-1, C)`` it — the
    reference's ``children()[:-1]`` contract
    (``/root/reference/main.py:190-193,238``).
    """

    def __init__(self, block: Type[Union[BasicBlock, Bottleneck]],
                 layers: List[int], in_channels: int = 3, groups: int = 1,
                 base_width: int = 64):
        super().__init__()
        self.groups = groups
        self.base_width = base_width
        self.inplanes = 64
        self.conv1 = nn.Conv2d(in_channels, 64, kernel_size=7, stride=2,
                               padding=3, bias=False)
        self.bn1 = FusedBatchNorm(64, relu=True)
        self.maxpool = nn.MaxPool2d(kernel_size=3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.out_channels = 512 * block.expansion

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, FusedBatchNorm):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

    def _make_layer(self, block, planes: int, blocks: int,
                    stride: int = 1) -> nn.Sequential:
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = Downsample(self.inplanes,
                                    planes * block.expansion, stride)
        kw = {}
        if block is Bottleneck:
            kw = dict(groups=self.groups, base_width=self.base_width)
        layers = [block(self.inplanes, planes, stride, downsample, **kw)]
        self.inplanes = planes * block.expansion
        layers += [block(self.inplanes, planes, **kw)
                   for _ in range(1, blocks)]
        return nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.bn1(self.conv1(x))
        x = self.maxpool(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        return self.avgpool(x)


def build_encoder(arch: str, in_channels: int = 3) -> ResNetEncoder:
    if arch not in ARCH_SPECS:
        raise ValueError(
            f"unknown arch {arch!r}; available: {', '.join(arch_names())}")
    block, layers, groups, base_width = ARCH_SPECS[arch]
    return ResNetEncoder(block, layers, in_channels=in_channels,
                         groups=groups, base_width=base_width)
