"""ResNet family — the workloads of both reference examples.

The MXNet example trains ``train_cifar10.py --network resnet`` (CIFAR
ResNets: 3×3 convs at 32²/16²/8², C=16/32/64, depth 6n+2) and the
TF/Horovod example trains ImageNet ResNet-50.
Both are rebuilt here natively:

* ``--num-layers 20/32/56/110`` → CIFAR basic-block ResNet (He et al.);
* ``--num-layers 18/34/50/101/152`` → ImageNet ResNet (bottleneck for
  >=50), the BASELINE.json headline model.

MI355X-first details: NHWC (`channels_last`) is the native layout of the
HIP conv kernels (ops/), so modules are layout-agnostic and the trainer
moves the model+batch to ``channels_last``; BatchNorm+ReLU pairs route
through the fused HIP op when the native impl is selected.
"""

from __future__ import annotations

from typing import List, Optional, Type, Union

import torch
import torch.nn as nn

from ..ops import functional as F_amd


class DLXConv2d(nn.Conv2d):
    """Conv2d dispatching to the implicit-GEMM MFMA kernel (ops/conv.hip).

    Falls back to the library path for shapes outside the native surface
    (Cin%8!=0 and Cin>4 — the 3-channel stems use the packed small-Cin
    path) or when DLX_CONV=torch.
    """

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F_amd.conv2d(x, self.weight, self.stride[0], self.padding[0])


def conv3x3(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    return DLXConv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def conv1x1(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    return DLXConv2d(cin, cout, 1, stride=stride, bias=False)


class StemMaxPool(nn.Module):
    """3x3/s2/p1 maxpool (native HIP NHWC kernel with argmax backward)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F_amd.maxpool3x3s2(x)


class GlobalAvgPool(nn.Module):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F_amd.global_avg_pool(x)


class ConvBN(nn.Module):
    """Downsample path: 1x1 conv + BN (no relu), fused on the native path."""

    def __init__(self, cin: int, cout: int, stride: int):
        super().__init__()
        self.conv = conv1x1(cin, cout, stride)
        self.bn = nn.BatchNorm2d(cout)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F_amd.conv_bn_act(x, self.conv.weight, self.bn,
                                 self.conv.stride[0], 0, relu=False,
                                 training=self.training)


class BNReLU(nn.Module):
    """BatchNorm2d + ReLU, fused on the native path (ops/bn_relu kernel)."""

    def __init__(self, channels: int, relu: bool = True):
        super().__init__()
        self.bn = nn.BatchNorm2d(channels)
        self.relu = relu

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F_amd.batch_norm_act(
            x, self.bn, relu=self.relu, training=self.training)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin: int, cout: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None):
        super().__init__()
        self.conv1 = conv3x3(cin, cout, stride)
        self.bn1 = BNReLU(cout, relu=True)
        self.conv2 = conv3x3(cout, cout)
        self.bn2 = BNReLU(cout, relu=False)
        self.downsample = downsample

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x if self.downsample is None else self.downsample(x)
        out = F_amd.conv_bn_act(x, self.conv1.weight, self.bn1.bn,
                                self.conv1.stride[0], self.conv1.padding[0],
                                relu=True, training=self.training)
        return F_amd.conv_bn_act(out, self.conv2.weight, self.bn2.bn,
                                 self.conv2.stride[0], self.conv2.padding[0],
                                 residual=identity, training=self.training)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, cout: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None):
        super().__init__()
        self.conv1 = conv1x1(cin, cout)
        self.bn1 = BNReLU(cout, relu=True)
        self.conv2 = conv3x3(cout, cout, stride)
        self.bn2 = BNReLU(cout, relu=True)
        self.conv3 = conv1x1(cout, cout * self.expansion)
        self.bn3 = BNReLU(cout * self.expansion, relu=False)
        self.downsample = downsample

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # one autograd node for the whole block on the native path: the
        # residual fan-in add is fused into conv1's dgrad epilogue
        ds = self.downsample
        return F_amd.bottleneck(
            x, self.conv1.weight, self.bn1.bn,
            self.conv2.weight, self.bn2.bn,
            self.conv2.stride[0], self.conv2.padding[0],
            self.conv3.weight, self.bn3.bn,
            down_convw=ds.conv.weight if ds is not None else None,
            down_bn=ds.bn if ds is not None else None,
            down_stride=ds.conv.stride[0] if ds is not None else 1,
            training=self.training)


class ResNet(nn.Module):
    def __init__(self, block: Type[Union[BasicBlock, Bottleneck]],
                 layers: List[int], num_classes: int,
                 stem: str = "imagenet", width: int = 64):
        super().__init__()
        self.stem_kind = stem
        self.inplanes = width if stem == "imagenet" else 16
        if stem == "imagenet":
            self.conv1 = DLXConv2d(3, self.inplanes, 7, stride=2, padding=3,
                                   bias=False)
            self.maxpool = StemMaxPool()
        else:  # cifar: single 3x3 stem, no maxpool
            self.conv1 = conv3x3(3, self.inplanes)
            self.maxpool = nn.Identity()
        self.bn1 = BNReLU(self.inplanes, relu=True)

        stages = []
        planes = self.inplanes
        for i, n in enumerate(layers):
            stages.append(self._make_stage(block, planes * (2 ** i), n,
                                           stride=1 if i == 0 else 2))
        self.stages = nn.Sequential(*stages)
        self.avgpool = GlobalAvgPool()
        self.fc = nn.Linear(self.inplanes, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        # zero-init last BN gamma of each block (standard large-batch trick)
        for m in self.modules():
            if isinstance(m, Bottleneck):
                nn.init.zeros_(m.bn3.bn.weight)
            elif isinstance(m, BasicBlock):
                nn.init.zeros_(m.bn2.bn.weight)

    def _make_stage(self, block, planes: int, blocks: int,
                    stride: int) -> nn.Sequential:
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = ConvBN(self.inplanes, planes * block.expansion,
                                stride)
        layers = [block(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(self.inplanes, planes))
        return nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.bn1(self.conv1(x))
        x = self.maxpool(x)
        x = self.stages(x)
        x = self.avgpool(x)
        return self.fc(torch.flatten(x, 1))


_IMAGENET_DEPTHS = {
    18: (BasicBlock, [2, 2, 2, 2]),
    34: (BasicBlock, [3, 4, 6, 3]),
    50: (Bottleneck, [3, 4, 6, 3]),
    101: (Bottleneck, [3, 4, 23, 3]),
    152: (Bottleneck, [3, 8, 36, 3]),
}


def build_resnet(num_layers: int = 50, num_classes: Optional[int] = None,
                 dataset: str = "imagenet") -> ResNet:
    """``--network resnet --num-layers N`` entry point.

    ``dataset='cifar10'`` selects the CIFAR stem and depth rule 6n+2
    (20, 32, 44, 56, 110 …); ``'imagenet'`` the standard depth table.
    """
    if dataset.startswith("cifar"):
        if (num_layers - 2) % 6 != 0:
            raise ValueError(f"CIFAR ResNet depth must be 6n+2, got {num_layers}")
        n = (num_layers - 2) // 6
        return ResNet(BasicBlock, [n, n, n],
                      num_classes=num_classes or 10, stem="cifar", width=16)
    if num_layers not in _IMAGENET_DEPTHS:
        raise ValueError(f"unsupported ImageNet depth {num_layers}; "
                         f"choose {sorted(_IMAGENET_DEPTHS)}")
    block, layers = _IMAGENET_DEPTHS[num_layers]
    return ResNet(block, layers, num_classes=num_classes or 1000,
                  stem="imagenet", width=64)
