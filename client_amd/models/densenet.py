"""DenseNet-121 (from scratch on torch.nn).

BASELINE.md config 3 names densenet_onnx (the canonical Triton qa
model) for the 8-replica concurrency sweep; this is the same
architecture served natively (random-init, synthetic inputs).
"""

import torch
import torch.nn as nn


class DenseLayer(nn.Module):
    def __init__(self, in_ch, growth, bn_size=4):
        super().__init__()
        self.norm1 = nn.BatchNorm2d(in_ch)
        self.conv1 = nn.Conv2d(in_ch, bn_size * growth, 1, bias=False)
        self.norm2 = nn.BatchNorm2d(bn_size * growth)
        self.conv2 = nn.Conv2d(bn_size * growth, growth, 3, padding=1,
                               bias=False)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        out = self.conv1(self.relu(self.norm1(x)))
        out = self.conv2(self.relu(self.norm2(out)))
        return torch.cat([x, out], 1)


class Transition(nn.Module):
    def __init__(self, in_ch, out_ch):
        super().__init__()
        self.norm = nn.BatchNorm2d(in_ch)
        self.conv = nn.Conv2d(in_ch, out_ch, 1, bias=False)
        self.pool = nn.AvgPool2d(2, stride=2)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        return self.pool(self.conv(self.relu(self.norm(x))))


class DenseNet121(nn.Module):
    def __init__(self, growth=32, blocks=(6, 12, 24, 16), num_classes=1000):
        super().__init__()
        ch = 64
        self.stem = nn.Sequential(
            nn.Conv2d(3, ch, 7, stride=2, padding=3, bias=False),
            nn.BatchNorm2d(ch), nn.ReLU(inplace=True),
            nn.MaxPool2d(3, stride=2, padding=1),
        )
        layers = []
        for i, n in enumerate(blocks):
            for _ in range(n):
                layers.append(DenseLayer(ch, growth))
                ch += growth
            if i != len(blocks) - 1:
                layers.append(Transition(ch, ch // 2))
                ch //= 2
        self.features = nn.Sequential(*layers)
        self.norm = nn.BatchNorm2d(ch)
        self.classifier = nn.Linear(ch, num_classes)

    def forward(self, x):
        x = self.features(self.stem(x))
        x = torch.relu(self.norm(x))
        x = torch.nn.functional.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.classifier(x)


def densenet121():
    return DenseNet121()
