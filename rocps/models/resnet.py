"""ResNet-50 / Wide-ResNet-101-2 — the benchmark models of
BASELINE.json configs[1,2,4] (the reference exercises them through
TF's Conv2D/FusedBatchNorm/Relu/MaxPool kernel stack).

MI355X-first choices:
- NHWC (channels_last) activations end-to-end: implicit-GEMM conv on
  CDNA4 wants C-minor layouts (MIOpen picks its fastest paths there,
  and our fused kernels index C contiguously).
- bf16 compute with fp32 master weights in the optimizer (C8).
- BN+ReLU run as one fused HIP kernel per (train fwd) call on GPU
  (rocps.ops.norm; csrc/kernels/bn_relu.hip) — two HBM passes instead
  of four (BN stats + normalize+scale+relu, vs conv-out read, BN, relu,
  write each separately).
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

from rocps.ops.conv1x1 import Conv1x1
from rocps.ops.norm import BatchNormAct2d
from rocps.ops.pool import MaxPool332

__all__ = ["ResNet", "resnet50", "wide_resnet101_2"]


def conv3x3(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def conv1x1(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    # Conv1x1 runs the MFMA-GEMM train path on shapes where it measured
    # faster than MIOpen (rocps/ops/conv1x1.py dispatch table);
    # state_dict-compatible with nn.Conv2d.
    return Conv1x1(cin, cout, stride=stride)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, planes: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None,
                 base_width: int = 64):
        super().__init__()
        width = int(planes * base_width / 64)
        self.conv1 = conv1x1(cin, width)
        self.bn1 = BatchNormAct2d(width, act="relu")
        self.conv2 = conv3x3(width, width, stride)
        self.bn2 = BatchNormAct2d(width, act="relu")
        self.conv3 = conv1x1(width, planes * self.expansion)
        # bn3 fuses the skip-add + final ReLU into its epilogue (C9)
        self.bn3 = BatchNormAct2d(planes * self.expansion, act="add_relu")
        self.downsample = downsample

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.bn3(self.conv3(out), residual=identity)


class ResNet(nn.Module):
    def __init__(self, layers: List[int], num_classes: int = 1000,
                 base_width: int = 64):
        super().__init__()
        self.inplanes = 64
        self.base_width = base_width
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = BatchNormAct2d(64, act="relu")
        self.maxpool = MaxPool332()
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], stride=2)
        self.layer3 = self._make_layer(256, layers[2], stride=2)
        self.layer4 = self._make_layer(512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
        # zero-init last BN gamma per block (standard recipe)
        for m in self.modules():
            if isinstance(m, Bottleneck):
                nn.init.zeros_(m.bn3.weight)

    def _make_layer(self, planes: int, blocks: int, stride: int = 1):
        downsample = None
        if stride != 1 or self.inplanes != planes * Bottleneck.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * Bottleneck.expansion, stride),
                BatchNormAct2d(planes * Bottleneck.expansion, act="none"),
            )
        layers = [Bottleneck(self.inplanes, planes, stride, downsample,
                             self.base_width)]
        self.inplanes = planes * Bottleneck.expansion
        layers += [Bottleneck(self.inplanes, planes,
                              base_width=self.base_width)
                   for _ in range(1, blocks)]
        return nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)

    def loss(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        from rocps.ops.softmax_xent import softmax_cross_entropy_with_logits
        return softmax_cross_entropy_with_logits(self(x), y).mean()


def resnet50(num_classes: int = 1000, **kw) -> ResNet:
    return ResNet([3, 4, 6, 3], num_classes=num_classes, **kw)


def wide_resnet101_2(num_classes: int = 1000, **kw) -> ResNet:
    return ResNet([3, 4, 23, 3], num_classes=num_classes, base_width=128, **kw)
