"""ResNet v1 (50/101/152) — the dense AR-mode benchmark workload.

Reference: examples/tf_cnn_benchmarks/models/resnet_model.py:42-258
(bottleneck v1 with projection shortcuts, BN decay .999, 1001-way softmax,
224x224 synthetic ImageNet, per-GPU batch 64) driven by
benchmark_cnn.py (SGD/Momentum, softmax-CE loss :290-302).

MI355X-native choices: NHWC (channels_last) memory format — the fast
layout for MIOpen convolutions on CDNA4 (the reference defaults to NCHW
for CUDA); bf16 autocast for the conv/GEMM path; our fused
softmax-CE kernel for the loss; all parameters dense -> the whole model
exercises the bucketed RCCL all-reduce engine.
"""

import os as _os

import torch
import torch.nn as nn

from parallax_amd import optim
from parallax_amd.ops.softmax_ce import softmax_cross_entropy
from parallax_amd.session import TrainGraph

_LAYERS = {50: (3, 4, 6, 3), 101: (3, 4, 23, 3), 152: (3, 8, 36, 3)}


def _bn(c, bn_momentum, relu_after=False):
    """Hand-written NHWC BN with the following ReLU fused (default;
    ops/batchnorm.py), or nn.BatchNorm2d with PARALLAX_NATIVE_BN=0."""
    if _bn_fused():
        from parallax_amd.ops.batchnorm import ParallaxBatchNorm2d
        return ParallaxBatchNorm2d(c, momentum=bn_momentum,
                                   relu=relu_after)
    return nn.BatchNorm2d(c, momentum=bn_momentum)


def _bn_fused():
    # default ON: the fused NHWC BN(+ReLU)(+residual add) kernels remove
    # the MIOpen BN + elementwise glue that measured ~50% of the tuned
    # ResNet step (gpurun_out/prof_rn50, round 2)
    return _os.environ.get('PARALLAX_NATIVE_BN', '1') == '1'


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, width, stride=1, bn_momentum=0.001):
        super().__init__()
        cout = width * self.expansion
        self.conv1 = nn.Conv2d(cin, width, 1, bias=False)
        self.bn1 = _bn(width, bn_momentum, relu_after=True)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = _bn(width, bn_momentum, relu_after=True)
        self.conv3 = nn.Conv2d(width, cout, 1, bias=False)
        # fused mode: bn3 also folds the shortcut add + final ReLU
        self.bn3 = _bn(cout, bn_momentum, relu_after=True) \
            if _bn_fused() else _bn(cout, bn_momentum)
        self.fused_bn3 = _bn_fused()
        self.relu = nn.ReLU(inplace=True)
        # with fused BN+ReLU the separate activation after bn1/bn2 is a
        # no-op; keep it as Identity to preserve the module structure
        self.act = nn.Identity() if _bn_fused() else self.relu
        if stride != 1 or cin != cout:
            self.downsample = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride=stride, bias=False),
                _bn(cout, bn_momentum))
        else:
            self.downsample = None

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = self.act(self.bn1(self.conv1(x)))
        out = self.act(self.bn2(self.conv2(out)))
        if self.fused_bn3:
            return self.bn3(self.conv3(out), identity)
        out = self.bn3(self.conv3(out))
        return self.relu(out + identity)


class BottleneckV2(nn.Module):
    """Pre-activation bottleneck (ResNet v2, reference
    resnet_model.py:129-180)."""

    expansion = 4

    def __init__(self, cin, width, stride=1, bn_momentum=0.001):
        super().__init__()
        cout = width * self.expansion
        self.bn1 = nn.BatchNorm2d(cin, momentum=bn_momentum)
        self.conv1 = nn.Conv2d(cin, width, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(width, momentum=bn_momentum)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn3 = nn.BatchNorm2d(width, momentum=bn_momentum)
        self.conv3 = nn.Conv2d(width, cout, 1, bias=False)
        self.relu = nn.ReLU(inplace=True)
        if stride != 1 or cin != cout:
            self.downsample = nn.Conv2d(cin, cout, 1, stride=stride,
                                        bias=False)
        else:
            self.downsample = None

    def forward(self, x):
        pre = self.relu(self.bn1(x))
        identity = x if self.downsample is None else self.downsample(pre)
        out = self.conv1(pre)
        out = self.conv2(self.relu(self.bn2(out)))
        out = self.conv3(self.relu(self.bn3(out)))
        return out + identity


class ResNet(nn.Module):
    def __init__(self, depth=50, num_classes=1001, bn_momentum=0.001,
                 version=1):
        super().__init__()
        self.version = version
        layers = _LAYERS[depth]
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = _bn(64, bn_momentum, relu_after=True) \
            if version == 1 else nn.BatchNorm2d(64, momentum=bn_momentum)
        self.stem_fused = _bn_fused() and version == 1
        self.relu = nn.ReLU(inplace=True)
        if _os.environ.get('PARALLAX_NATIVE_POOL', '1') == '1':
            from parallax_amd.ops.pool import ParallaxMaxPool3x3s2
            self.maxpool = ParallaxMaxPool3x3s2()
        else:
            self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        block = Bottleneck if version == 1 else BottleneckV2
        cin, stages = 64, []
        for i, (width, n) in enumerate(zip((64, 128, 256, 512), layers)):
            blocks = []
            for j in range(n):
                stride = 2 if (i > 0 and j == 0) else 1
                blocks.append(block(cin, width, stride, bn_momentum))
                cin = width * block.expansion
            stages.append(nn.Sequential(*blocks))
        self.layer1, self.layer2, self.layer3, self.layer4 = stages
        if version == 2:
            self.bn_final = nn.BatchNorm2d(cin, momentum=bn_momentum)
        if _os.environ.get('PARALLAX_NATIVE_POOL', '1') == '1':
            from parallax_amd.ops.pool import ParallaxGlobalAvgPool
            self.avgpool = ParallaxGlobalAvgPool()
            self._flat_pool = True
        else:
            self.avgpool = nn.AdaptiveAvgPool2d(1)
            self._flat_pool = False
        self.fc = nn.Linear(512 * block.expansion, num_classes)

    def forward(self, x):
        x = self.bn1(self.conv1(x))
        if not self.stem_fused:
            x = self.relu(x)
        x = self.maxpool(x)
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        if self.version == 2:
            x = self.relu(self.bn_final(x))
        x = self.avgpool(x)
        if not self._flat_pool:
            x = x.flatten(1)
        return self.fc(x.to(self.fc.weight.dtype))


def build_resnet_graph(depth=50, num_classes=1001, batch_size=64, lr=0.1,
                       momentum=0.9, weight_decay=1e-4, use_bf16=True,
                       channels_last=True, pure_bf16=False):
    """pure_bf16: conv/FC weights live in bf16 (no per-step autocast
    weight casts; fp32 update precision via the fused master-weight
    optimizer path); BatchNorm stays fp32 (mixed bf16-input BN is the
    library's standard mixed-precision path)."""
    model = ResNet(depth, num_classes)
    import os as _os
    if _os.environ.get('PARALLAX_NATIVE_CONV1X1') == '1' and channels_last:
        from parallax_amd.ops.conv import swap_conv1x1
        swap_conv1x1(model)
    if _os.environ.get('PARALLAX_NATIVE_CONV3X3') == '1' and channels_last:
        from parallax_amd.ops.conv import swap_conv3x3
        swap_conv3x3(model)
    if channels_last:
        model = model.to(memory_format=torch.channels_last)
    if pure_bf16:
        import torch.nn as _nn
        from parallax_amd.ops.batchnorm import ParallaxBatchNorm2d
        model = model.to(torch.bfloat16)
        for mod in model.modules():
            if isinstance(mod, (_nn.modules.batchnorm._BatchNorm,
                                ParallaxBatchNorm2d)):
                mod.float()
    optimizer = optim.SGD(model.parameters(), lr=lr, momentum=momentum,
                          weight_decay=weight_decay)
    use_amp = use_bf16 and not pure_bf16

    def step_fn(m, batch):
        x, y = batch
        if channels_last and x.dim() == 4:
            x = x.to(memory_format=torch.channels_last)
        if pure_bf16:
            logits = m(x.to(torch.bfloat16))
            loss = softmax_cross_entropy(logits.float(), y)
        elif use_amp and x.is_cuda:
            with torch.autocast('cuda', dtype=torch.bfloat16):
                logits = m(x)
            loss = softmax_cross_entropy(logits.float(), y)
        else:
            logits = m(x)
            loss = softmax_cross_entropy(logits, y)
        return loss, {}

    return TrainGraph(model, optimizer, step_fn)


def synthetic_batch(batch_size=64, image_size=224, num_classes=1001,
                    device='cpu', generator=None):
    """Synthetic ImageNet batch (reference benchmark_cnn.py:836-860)."""
    x = torch.randn(batch_size, 3, image_size, image_size, device=device,
                    generator=generator)
    y = torch.randint(0, num_classes, (batch_size,), device=device,
                      generator=generator)
    return x, y
