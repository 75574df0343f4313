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

        if x.is_cuda and x.dtype == torch.bfloat16:
            from ..ops import hip_runtime as hr

            if self.bias.dtype != torch.float32:
                # TorchModel's .to(bf16) sweeps buffers too; the kernel
                # reads fp32 bias — normalize once (pre-capture, the
                # eager warmup hits this before any hipGraph capture)
                self.bias.data = self.bias.data.float()

            c = x.shape[1]
            if x.is_contiguous():
                n = x.shape[0]
                plane = x.numel() // (n * c)
                hr.bias_act_bf16(
                    x.data_ptr(), self.bias.data_ptr(), x.data_ptr(),
                    n * c, plane, c, self.relu_flag,
                    torch.cuda.current_stream().cuda_stream)
                return x
            if (x.dim() == 4 and c % 8 == 0
                    and x.is_contiguous(
                        memory_format=torch.channels_last)):
                # NHWC memory: channel is the fastest dim — the cl
                # kernel keeps the fusion on MIOpen's native layout
                hr.bias_res_act_cl_bf16(
                    x.data_ptr(), 0, self.bias.data_ptr(), x.data_ptr(),
                    x.numel(), c, self.relu_flag,
                    torch.cuda.current_stream().cuda_stream)
                return x
        out = x + self.bias.view(1, -1, *([1] * (x.dim() - 2))).to(x.dtype)
        return out.relu_() if self.relu_flag else out


class BiasResAct(nn.Module):
    """Bottleneck-exit fusion: out = relu(x + bias[c] + residual) in ONE
    CDNA4 kernel pass. After BN folding torch-on-ROCm runs this as three
    separate elementwise kernels (bias add, residual add, relu) — three
    full memory passes over the activation per block
    (profiles/serving_rocprof_postfold_r01.txt). fp32 math in-kernel,
    RNE back to bf16."""

    def __init__(self, bias):
        super().__init__()
        self.register_buffer("bias", bias.detach().float())

    def forward(self, x, residual):
        import torch

        if (x.is_cuda and x.dtype == torch.bfloat16
                and residual.dtype == torch.bfloat16):
            from ..ops import hip_runtime as hr

            if self.bias.dtype != torch.float32:
                # TorchModel's .to(bf16) sweeps buffers; kernel reads fp32
                self.bias.data = self.bias.data.float()
            c = x.shape[1]
            if x.is_contiguous() and residual.is_contiguous():
                n = x.shape[0]
                plane = x.numel() // (n * c)
                hr.bias_res_act_bf16(
                    x.data_ptr(), residual.data_ptr(),
                    self.bias.data_ptr(), x.data_ptr(), n * c, plane, c,
                    True, torch.cuda.current_stream().cuda_stream)
                return x
            cl = torch.channels_last
            if (x.dim() == 4 and c % 8 == 0
                    and x.is_contiguous(memory_format=cl)
                    and residual.is_contiguous(memory_format=cl)):
                hr.bias_res_act_cl_bf16(
                    x.data_ptr(), residual.data_ptr(),
                    self.bias.data_ptr(), x.data_ptr(), x.numel(), c,
                    True, torch.cuda.current_stream().cuda_stream)
                return x
        out = x + residual + self.bias.view(
            1, -1, *([1] * (x.dim() - 2))
        ).to(x.dtype)
        return out.relu_()


class FusedBottleneck(nn.Module):
    """Bottleneck with BN folded into the convs and every elementwise
    epilogue fused into one-pass CDNA4 kernels:

        conv1 -> BiasAct(relu)     1 pass  (was bias add + relu = 2)
        conv2 -> BiasAct(relu)     1 pass
        conv3 (biasless), identity = downsample conv (biasless) or x
              -> BiasResAct        1 pass  (was bias [+ds bias] + add
                                            + relu = 3-4 passes)

    The downsample conv's folded bias is per-channel on the same C as
    conv3's, so it is pre-summed into the exit bias."""

    def __init__(self, conv1, ba1, conv2, ba2, conv3, ds_conv, exit_bias):
        super().__init__()
        self.conv1, self.ba1 = conv1, ba1
        self.conv2, self.ba2 = conv2, ba2
        self.conv3 = conv3
        self.ds_conv = ds_conv
        self.exit = BiasResAct(exit_bias)

    def forward(self, x):
        identity = self.ds_conv(x) if self.ds_conv is not None else x
        out = self.ba1(self.conv1(x))
        out = self.ba2(self.conv2(out))
        out = self.conv3(out)
        return self.exit(out, identity)


def _strip_bias(conv):
    bias = conv.bias.detach().clone()
    conv.bias = None
    return bias


def fold_batchnorm(module, fuse_eltwise=None):
    """Fold every Conv2d -> BatchNorm2d pair into the conv weights
    (algebraically exact in eval mode: W' = W*g/sqrt(v+eps),
    b' = b0*g/sqrt(v+eps) + beta - g*mean/sqrt(v+eps)). The serving
    profile showed BN kernels at ~29% of ResNet50 kernel time
    (profiles/serving_rocprof_r01.txt); folding removes them entirely.

    fuse_eltwise=True additionally rewrites every Bottleneck as a
    FusedBottleneck and the stem bias+relu as one BiasAct pass, so the
    folded conv biases never run as separate torch elementwise kernels
    (default; CLIENT_AMD_FUSED_BIAS=0 disables). Pairs are detected by
    _modules adjacency (conv immediately followed by its BN, matching
    this file and torchvision layouts)."""
    from torch.nn.utils.fusion import fuse_conv_bn_eval

    if fuse_eltwise is None:
        fuse_eltwise = os.environ.get("CLIENT_AMD_FUSED_BIAS", "1") != "0"

    def fuse_pair(conv, bn):
        return fuse_conv_bn_eval(conv.eval(), bn.eval())

    if fuse_eltwise:
        # bottlenecks -> FusedBottleneck
        for m in list(module.modules()):
            for name, block in list(m._modules.items()):
                if not isinstance(block, Bottleneck):
                    continue
                c1 = fuse_pair(block.conv1, block.bn1)
                c2 = fuse_pair(block.conv2, block.bn2)
                c3 = fuse_pair(block.conv3, block.bn3)
                ba1 = BiasAct(_strip_bias(c1), relu=True)
                ba2 = BiasAct(_strip_bias(c2), relu=True)
                exit_bias = _strip_bias(c3)
                ds_conv = None
                if block.downsample is not None:
                    ds_conv = fuse_pair(block.downsample[0],
                                        block.downsample[1])
                    exit_bias = exit_bias + _strip_bias(ds_conv)
                m._modules[name] = FusedBottleneck(
                    c1, ba1, c2, ba2, c3, ds_conv, exit_bias
                )
        # stem: conv1+bn1+relu -> biasless conv + one BiasAct(relu) pass
        if (isinstance(getattr(module, "conv1", None), nn.Conv2d)
                and isinstance(getattr(module, "bn1", None),
                               nn.BatchNorm2d)):
            fused = fuse_pair(module.conv1, module.bn1)
            module.conv1 = fused
            module.bn1 = BiasAct(_strip_bias(fused), relu=True)
            module.relu = nn.Identity()

    # generic pass folds any remaining conv->bn adjacency (stem when
    # fuse_eltwise is off, custom models)
    for m in list(module.modules()):
        names = list(m._modules.keys())
        for a, b in zip(names, names[1:]):
            conv, bn = m._modules[a], m._modules[b]
            if isinstance(conv, nn.Conv2d) and isinstance(bn, nn.BatchNorm2d):
                m._modules[a] = fuse_pair(conv, bn)
                m._modules[b] = nn.Identity()
    return module


def resnet50(fold_bn=True):
    model = ResNet50()
    if fold_bn:
        model = fold_batchnorm(model.eval())
    return model
