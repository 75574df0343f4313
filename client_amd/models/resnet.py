"""ResNet-50 (v1.5) for the benchmark server.

The BASELINE north-star metric is perf_analyzer-style inferences/sec on
ResNet50 bs=8 with HIP-shm I/O. torchvision is not in
this image, so the standard architecture is defined here directly
(random-init weights; the benchmark uses synthetic inputs).
"""

import os

import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, width, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, width * self.expansion, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(width * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        out += identity
        return self.relu(out)


class ResNet50(nn.Module):
    def __init__(self, num_classes=1000):
        super().__init__()
        self.in_ch = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, 3)
        self.layer2 = self._make_layer(128, 4, stride=2)
        self.layer3 = self._make_layer(256, 6, stride=2)
        self.layer4 = self._make_layer(512, 3, stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)

    def _make_layer(self, width, blocks, stride=1):
        downsample = None
        out_ch = width * Bottleneck.expansion
        if stride != 1 or self.in_ch != out_ch:
            downsample = nn.Sequential(
                nn.Conv2d(self.in_ch, out_ch, 1, stride=stride, bias=False),
                nn.BatchNorm2d(out_ch),
            )
        layers = [Bottleneck(self.in_ch, width, stride, downsample)]
        self.in_ch = out_ch
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.in_ch, width))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


class BiasAct(nn.Module):
    """Per-channel bias (+optional ReLU) as one CDNA4 kernel pass.

    torch-on-ROCm emits a separate elementwise kernel for conv bias
    (miopen_convolution has no bias epilogue), so after BN folding the
    bias costs a full extra memory pass. Round-2 lever, gated by
    CLIENT_AMD_FUSED_BIAS=1 (kernel numerics are GPU-tested; default
    path keeps bias on the conv)."""

    def __init__(self, bias, relu=False):
        super().__init__()
        self.register_buffer("bias", bias.detach().float())
        self.relu_flag = relu

    def forward(self, x):
        import torch

        if (x.is_cuda and x.dtype == torch.bfloat16
                and x.is_contiguous()):
            from ..ops import hip_runtime as hr

            if self.bias.dtype != torch.float32:
                # TorchModel's .to(bf16) sweeps buffers too; the kernel
                # reads fp32 bias — normalize once (pre-capture, the
                # eager warmup hits this before any hipGraph capture)
                self.bias.data = self.bias.data.float()

            n, c = x.shape[0], x.shape[1]
            plane = x.numel() // (n * c)
            hr.bias_act_bf16(
                x.data_ptr(), self.bias.data_ptr(), x.data_ptr(),
                n * c, plane, c, self.relu_flag,
                torch.cuda.current_stream().cuda_stream)
            return x
        out = x + self.bias.view(1, -1, *([1] * (x.dim() - 2))).to(x.dtype)
        return out.relu_() if self.relu_flag else out


def fold_batchnorm(module):
    """Fold every Conv2d -> BatchNorm2d pair into the conv weights
    (algebraically exact in eval mode: W' = W*g/sqrt(v+eps),
    b' = b0*g/sqrt(v+eps) + beta - g*mean/sqrt(v+eps)). The serving
    profile showed BN kernels at ~29% of ResNet50 kernel time
    (profiles/serving_rocprof_r01.txt); folding removes them entirely.
    Pairs are detected by _modules adjacency (conv immediately followed
    by its BN, which matches this file and torchvision layouts); the BN
    slot is replaced with Identity so forwards run unchanged."""
    from torch.nn.utils.fusion import fuse_conv_bn_eval

    for m in list(module.modules()):
        names = list(m._modules.keys())
        for a, b in zip(names, names[1:]):
            conv, bn = m._modules[a], m._modules[b]
            if isinstance(conv, nn.Conv2d) and isinstance(bn, nn.BatchNorm2d):
                fused = fuse_conv_bn_eval(conv.eval(), bn.eval())
                if os.environ.get("CLIENT_AMD_FUSED_BIAS") == "1":
                    # bias as one fused kernel pass instead of torch's
                    # separate elementwise add (see BiasAct)
                    bias = fused.bias.detach().clone()
                    fused.bias = None
                    m._modules[a] = fused
                    m._modules[b] = BiasAct(bias, relu=False)
                else:
                    m._modules[a] = fused
                    m._modules[b] = nn.Identity()
    return module


def resnet50(fold_bn=True):
    model = ResNet50()
    if fold_bn:
        model = fold_batchnorm(model.eval())
    return model
