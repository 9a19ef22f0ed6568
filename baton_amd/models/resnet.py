"""ResNet-18/50 (CIFAR-shaped), NHWC, built entirely on the HIP op layer.

BASELINE.json configs 2 and 5: the headline federated benchmark model.
Every hot op routes through baton_amd.ops — implicit-GEMM conv, fused
BN+ReLU, fused residual add+ReLU, MFMA Linear head, fused CE loss. The only
torch glue is the global average pool (a tiny [N,HW,C] mean) and tensor
reshapes.

CIFAR variant (32x32 inputs): 3x3 stem, no maxpool — the standard CIFAR
ResNet. Input layout NHWC: [N, 32, 32, 3].
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from baton_amd.ops import functional as BF
from baton_amd.ops.modules import (
    BatonBatchNorm2d,
    BatonConv2d,
    BatonLinear,
)
from baton_amd.runtime.local import LocalTrainer
from baton_amd.utils.config import TrainConfig


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin: int, cout: int, stride: int = 1):
        super().__init__()
        self.conv1 = BatonConv2d(cin, cout, 3, stride, 1)
        self.bn1 = BatonBatchNorm2d(cout, fused_relu=True)
        self.conv2 = BatonConv2d(cout, cout, 3, 1, 1)
        self.bn2 = BatonBatchNorm2d(cout)
        self.downsample: Optional[nn.Module] = None
        if stride != 1 or cin != cout:
            self.downsample = nn.Sequential(
                BatonConv2d(cin, cout, 1, stride, 0),
                BatonBatchNorm2d(cout),
            )

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))        # fused ReLU
        # residual join fused into bn2's normalize pass
        return self.bn2.forward_add_relu(self.conv2(out), identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, width: int, stride: int = 1):
        super().__init__()
        cout = width * self.expansion
        self.conv1 = BatonConv2d(cin, width, 1, 1, 0)
        self.bn1 = BatonBatchNorm2d(width, fused_relu=True)
        self.conv2 = BatonConv2d(width, width, 3, stride, 1)
        self.bn2 = BatonBatchNorm2d(width, fused_relu=True)
        self.conv3 = BatonConv2d(width, cout, 1, 1, 0)
        self.bn3 = BatonBatchNorm2d(cout)
        self.downsample: Optional[nn.Module] = None
        if stride != 1 or cin != cout:
            self.downsample = nn.Sequential(
                BatonConv2d(cin, cout, 1, stride, 0),
                BatonBatchNorm2d(cout),
            )

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3.forward_add_relu(self.conv3(out), identity)


class ResNet(nn.Module):
    def __init__(self, block, layers: List[int], num_classes: int = 10,
                 train_config: Optional[TrainConfig] = None):
        super().__init__()
        self.stem_conv = BatonConv2d(3, 64, 3, 1, 1)
        self.stem_bn = BatonBatchNorm2d(64, fused_relu=True)
        self.cin = 64
        self.layer1 = self._make_layer(block, 64, layers[0], 1)
        self.layer2 = self._make_layer(block, 128, layers[1], 2)
        self.layer3 = self._make_layer(block, 256, layers[2], 2)
        self.layer4 = self._make_layer(block, 512, layers[3], 2)
        self.fc = BatonLinear(512 * block.expansion, num_classes)
        self._trainer = LocalTrainer(
            train_config or TrainConfig(), loss_fn=BF.cross_entropy
        )

    def _make_layer(self, block, width: int, n: int, stride: int):
        blocks = [block(self.cin, width, stride)]
        self.cin = width * block.expansion
        for _ in range(n - 1):
            blocks.append(block(self.cin, width, 1))
        return nn.Sequential(*blocks)

    def forward(self, x):
        # x: [N, H, W, 3] NHWC
        out = self.stem_bn(self.stem_conv(x))
        out = self.layer1(out)
        out = self.layer2(out)
        out = self.layer3(out)
        out = self.layer4(out)
        out = out.mean(dim=(1, 2))           # global average pool (glue)
        return self.fc(out)

    def train_round(self, *data, n_epoch: int = 1):
        return self._trainer(self, data, n_epoch)


def resnet18(num_classes: int = 10, train_config: Optional[TrainConfig] = None) -> ResNet:
    m = ResNet(BasicBlock, [2, 2, 2, 2], num_classes, train_config)
    m.name = "resnet18"
    return m


def resnet50(num_classes: int = 10, train_config: Optional[TrainConfig] = None) -> ResNet:
    m = ResNet(Bottleneck, [3, 4, 6, 3], num_classes, train_config)
    m.name = "resnet50"
    return m


def make_synthetic_cifar(
    n_samples: int, num_classes: int = 10, seed: int = 0,
    dtype: torch.dtype = torch.float32,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """CIFAR-shaped synthetic data (no network for real datasets): NHWC
    images N(0,1) and uniform labels."""
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n_samples, 32, 32, 3, generator=g).to(dtype)
    y = torch.randint(0, num_classes, (n_samples,), generator=g)
    return x, y
