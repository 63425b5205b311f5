"""ResNet-50 (v1.5) in plain PyTorch — the BASELINE config-2 model family.

Written from scratch (no torchvision in this environment); standard
bottleneck architecture, random-init weights, NCHW fp32/bf16 inference.
"""
from __future__ import annotations

import torch
import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch: int, width: int, stride: int = 1):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(out_ch)
        self.relu = nn.ReLU(inplace=True)
        if stride != 1 or in_ch != out_ch:
            self.downsample = nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 1, stride=stride, bias=False),
                nn.BatchNorm2d(out_ch))
        else:
            self.downsample = None

    def forward(self, x):
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.relu(out + identity)


class ResNet(nn.Module):
    def __init__(self, layers, num_classes: int = 1000):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.in_ch = 64
        self.layer1 = self._make_layer(64, layers[0], 1)
        self.layer2 = self._make_layer(128, layers[1], 2)
        self.layer3 = self._make_layer(256, layers[2], 2)
        self.layer4 = self._make_layer(512, layers[3], 2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)

    def _make_layer(self, width, blocks, stride):
        layers = [Bottleneck(self.in_ch, width, stride)]
        self.in_ch = width * Bottleneck.expansion
        for _ in range(blocks - 1):
            layers.append(Bottleneck(self.in_ch, width))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet50(num_classes: int = 1000) -> ResNet:
    return ResNet([3, 4, 6, 3], num_classes)


def resnet50_servable(device: str = "cpu", dtype=torch.float32,
                      seed: int = 0):
    """ResNet-50 Servable: input alias "images" (N,3,224,224), output
    "logits" (N,1000)."""
    from ..server import Servable

    torch.manual_seed(seed)
    model = resnet50().to(device=device, dtype=dtype).eval()

    @torch.no_grad()
    def fn(inputs):
        x = inputs["images"]
        if not isinstance(x, torch.Tensor):
            x = torch.as_tensor(x)
        x = x.to(device=device, dtype=dtype)
        return {"logits": model(x).float()}

    s = Servable(
        fn,
        signature={
            "method_name": "tensorflow/serving/predict",
            "inputs": {"images": (1, [-1, 3, 224, 224])},   # DT_FLOAT
            "outputs": {"logits": (1, [-1, 1000])},
        })
    s.module = model
    return s
