"""ResNet-50 feature encoder, written from scratch (no torchvision).

Matches the behavior of the reference's encoder (ref
network/monodepth2/resnet_encoder.py:63-108): ImageNet-normalizes the
input, runs a standard ResNet-50 bottleneck stack, and returns 5 feature
taps [conv1+bn+relu, layer1..layer4] with channels [64, 256, 512, 1024,
2048] at strides [2, 4, 8, 16, 32].

MI355X-first differences:
  * no classification head — the reference carried torchvision's unused
    ``fc`` (25.6M-param model incl. fc) and needed
    ``DDP(find_unused_parameters=True)`` for it (ref
    synthesis_task.py:108); dropping it gives a static graph.
  * the ImageNet mean/std normalize is a registered buffer, created on
    the model's device (the reference hard-coded cuda:0 tensors, ref
    resnet_encoder.py:88-91).
  * designed to run in channels_last (NHWC) memory format under bf16
    autocast — the layout MIOpen's implicit-GEMM MFMA convs want.
"""
from __future__ import annotations

from typing import List

import torch
import torch.nn as nn

from mine_amd.ops.bn import FusedBNAct
from mine_amd.ops.conv_general import Conv2dMFMA

_IMAGENET_MEAN = (0.485, 0.456, 0.406)
_IMAGENET_STD = (0.229, 0.224, 0.225)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch: int, planes: int, stride: int = 1,
                 downsample: nn.Module = None):
        super().__init__()
        out_ch = planes * self.expansion
        self.conv1 = Conv2dMFMA(in_ch, planes, 1, bias=False)
        self.bn1 = FusedBNAct(planes, act="relu")
        self.conv2 = Conv2dMFMA(planes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn2 = FusedBNAct(planes, act="relu")
        self.conv3 = Conv2dMFMA(planes, out_ch, 1, bias=False)
        # residual join fused into the bn3 epilogue: relu(bn3(conv3) + id)
        self.bn3 = FusedBNAct(out_ch, act="add_relu")
        self.downsample = downsample

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), identity)


class ResNetEncoder(nn.Module):
    """5-tap ResNet-50 encoder. num_ch_enc = [64, 256, 512, 1024, 2048]."""

    def __init__(self, num_layers: int = 50, zero_init_residual: bool = True):
        super().__init__()
        if num_layers != 50:
            raise ValueError("only ResNet-50 is supported (ref hard-sets 50, "
                             "synthesis_task.py:68)")
        blocks = (3, 4, 6, 3)
        self.num_ch_enc = [64, 256, 512, 1024, 2048]

        self.conv1 = Conv2dMFMA(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = FusedBNAct(64, act="relu")
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(kernel_size=3, stride=2, padding=1)

        self.in_ch = 64
        self.layer1 = self._make_layer(64, blocks[0], stride=1)
        self.layer2 = self._make_layer(128, blocks[1], stride=2)
        self.layer3 = self._make_layer(256, blocks[2], stride=2)
        self.layer4 = self._make_layer(512, blocks[3], stride=2)

        mean = torch.tensor(_IMAGENET_MEAN).view(1, 3, 1, 1)
        std = torch.tensor(_IMAGENET_STD).view(1, 3, 1, 1)
        self.register_buffer("img_mean", mean, persistent=False)
        self.register_buffer("img_std", std, persistent=False)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1.0)
                nn.init.constant_(m.bias, 0.0)
        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck):
                    nn.init.constant_(m.bn3.weight, 0.0)

    def _make_layer(self, planes: int, num_blocks: int, stride: int) -> nn.Sequential:
        downsample = None
        out_ch = planes * Bottleneck.expansion
        if stride != 1 or self.in_ch != out_ch:
            downsample = nn.Sequential(
                Conv2dMFMA(self.in_ch, out_ch, 1, stride=stride, bias=False),
                FusedBNAct(out_ch, act="none"),
            )
        layers = [Bottleneck(self.in_ch, planes, stride, downsample)]
        self.in_ch = out_ch
        for _ in range(1, num_blocks):
            layers.append(Bottleneck(self.in_ch, planes))
        return nn.Sequential(*layers)

    def forward(self, img: torch.Tensor) -> List[torch.Tensor]:
        """img: Bx3xHxW in [0,1]. Returns 5 taps at strides 2..32."""
        x = (img - self.img_mean) / self.img_std
        conv1_out = self.bn1(self.conv1(x))  # relu fused in bn1
        b1 = self.layer1(self.maxpool(conv1_out))
        b2 = self.layer2(b1)
        b3 = self.layer3(b2)
        b4 = self.layer4(b3)
        return [conv1_out, b1, b2, b3, b4]
