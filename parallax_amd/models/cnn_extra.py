"""Remaining tf_cnn_benchmarks model families: VGG-11, LeNet-5,
Inception-v4, and the CIFAR-10 set (ResNet-20/32/44/56/110 v1/v2,
DenseNet-40/100, AlexNet-cifar, Trivial-cifar).

Reference inventory: models/model_config.py:29-63 (main + cifar10
registries); configs from models/vgg_model.py:56-62 ([1,1,2,2,2]),
models/lenet_model.py:27-38, models/inception_model.py:131-220
(inception_module column specs), models/resnet_model.py:281-366
(ResnetCifar10Model + residual_block :40-83 zero-pad Plan-A shortcut),
models/densenet_model.py:28-103, models/alexnet_model.py:52-77.

Everything here is a plain dense model -> exercised through the same
bucketed RCCL all-reduce engine as the headline ResNet-50; NHWC
(channels_last) + bf16 autocast on GPU via build_cnn_graph.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F


def _cbr(cin, cout, k, s=1, p='same'):
    """conv + BN + ReLU with TF SAME/VALID padding semantics."""
    if p == 'same':
        kh, kw = (k, k) if isinstance(k, int) else k
        p = (kh // 2, kw // 2)
    elif p == 'valid':
        p = 0
    return nn.Sequential(
        nn.Conv2d(cin, cout, k, stride=s, padding=p, bias=False),
        nn.BatchNorm2d(cout, momentum=0.001), nn.ReLU(inplace=True))


class VGG11(nn.Module):
    """vgg_model.py:56-62 — conv counts [1, 1, 2, 2, 2]."""

    CFG = [64, 'M', 128, 'M', 256, 256, 'M', 512, 512, 'M', 512, 512, 'M']

    def __init__(self, num_classes=1001):
        super().__init__()
        layers, cin = [], 3
        for v in self.CFG:
            if v == 'M':
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers += [nn.Conv2d(cin, v, 3, padding=1),
                           nn.ReLU(inplace=True)]
                cin = v
        self.features = nn.Sequential(*layers)
        self.classifier = nn.Sequential(
            nn.Flatten(), nn.Linear(512 * 7 * 7, 4096),
            nn.ReLU(inplace=True), nn.Dropout(0.5), nn.Linear(4096, 4096),
            nn.ReLU(inplace=True), nn.Dropout(0.5),
            nn.Linear(4096, num_classes))

    def forward(self, x):
        return self.classifier(self.features(x))


class LeNet5(nn.Module):
    """lenet_model.py:27-38 — 28x28 input, TF MNIST-tutorial layout
    (3-channel synthetic images, as in the benchmark's data path)."""

    def __init__(self, num_classes=1001):
        super().__init__()
        self.net = nn.Sequential(
            nn.Conv2d(3, 32, 5, padding=2), nn.ReLU(inplace=True),
            nn.MaxPool2d(2, 2),
            nn.Conv2d(32, 64, 5, padding=2), nn.ReLU(inplace=True),
            nn.MaxPool2d(2, 2),
            nn.Flatten(), nn.Linear(64 * 7 * 7, 512),
            nn.ReLU(inplace=True), nn.Linear(512, num_classes))

    def forward(self, x):
        return self.net(x)


# --- Inception-v4 (inception_model.py:131-220 column specs) -----------------

class _IncCols(nn.Module):
    """inception_module equivalent: parallel columns concatenated on
    channels.  A ('share', j-prefix) column reuses the previous column's
    intermediate outputs (the split-branch heads in inception_v4_c)."""

    def __init__(self, cin, cols):
        super().__init__()
        self.cols = nn.ModuleList()
        self.n_share = []
        self.cout = 0
        prev_chans = None
        for spec in cols:
            ops, chans = [], []
            c = cin
            nsh = 0
            for i, op in enumerate(spec):
                if op[0] == 'share':
                    nsh += 1
                    c = prev_chans[i]
                    chans.append(c)
                    continue
                if op[0] == 'conv':
                    _, cout, kh, kw, *rest = op
                    s = rest[0] if rest else 1
                    mode = rest[1] if len(rest) > 1 else 'same'
                    ops.append(_cbr(c, cout, (kh, kw), s, mode))
                    c = cout
                elif op[0] == 'mpool':
                    _, k, s, mode = op
                    ops.append(nn.MaxPool2d(k, s,
                                            k // 2 if mode == 'same' else 0))
                elif op[0] == 'apool':
                    _, k, s, mode = op
                    ops.append(nn.AvgPool2d(k, s,
                                            k // 2 if mode == 'same' else 0))
                chans.append(c)
            self.cols.append(nn.ModuleList(ops))
            self.n_share.append(nsh)
            self.cout += c
            prev_chans = chans

    def forward(self, x):
        outs, prev_inter = [], []
        for col, nsh in zip(self.cols, self.n_share):
            inter = prev_inter[:nsh]
            h = inter[-1] if inter else x
            for op in col:
                h = op(h)
                inter.append(h)
            outs.append(h)
            prev_inter = inter
        return torch.cat(outs, 1)


def _inc4_a(cin):   # 384 -> 384
    return _IncCols(cin, [
        [('apool', 3, 1, 'same'), ('conv', 96, 1, 1)],
        [('conv', 96, 1, 1)],
        [('conv', 64, 1, 1), ('conv', 96, 3, 3)],
        [('conv', 64, 1, 1), ('conv', 96, 3, 3), ('conv', 96, 3, 3)]])


def _inc4_b(cin):   # 1024 -> 1024
    return _IncCols(cin, [
        [('apool', 3, 1, 'same'), ('conv', 128, 1, 1)],
        [('conv', 384, 1, 1)],
        [('conv', 192, 1, 1), ('conv', 224, 1, 7), ('conv', 256, 7, 1)],
        [('conv', 192, 1, 1), ('conv', 192, 1, 7), ('conv', 224, 7, 1),
         ('conv', 224, 1, 7), ('conv', 256, 7, 1)]])


def _inc4_c(cin):   # 1536 -> 1536 (split-branch heads via 'share')
    return _IncCols(cin, [
        [('apool', 3, 1, 'same'), ('conv', 256, 1, 1)],
        [('conv', 256, 1, 1)],
        [('conv', 384, 1, 1), ('conv', 256, 1, 3)],
        [('share',), ('conv', 256, 3, 1)],
        [('conv', 384, 1, 1), ('conv', 448, 1, 3), ('conv', 512, 3, 1),
         ('conv', 256, 3, 1)],
        [('share',), ('share',), ('share',), ('conv', 256, 1, 3)]])


class InceptionV4(nn.Module):
    """Inception-v4 (299x299 input; inception_model.py:167-220)."""

    def __init__(self, num_classes=1001):
        super().__init__()
        self.net = nn.Sequential(
            _cbr(3, 32, 3, 2, 'valid'), _cbr(32, 32, 3, 1, 'valid'),
            _cbr(32, 64, 3),
            _IncCols(64, [[('mpool', 3, 2, 'valid')],
                          [('conv', 96, 3, 3, 2, 'valid')]]),          # sa
            _IncCols(160, [
                [('conv', 64, 1, 1), ('conv', 96, 3, 3, 1, 'valid')],
                [('conv', 64, 1, 1), ('conv', 64, 7, 1),
                 ('conv', 64, 1, 7),
                 ('conv', 96, 3, 3, 1, 'valid')]]),                    # sb
            _IncCols(192, [[('conv', 192, 3, 3, 2, 'valid')],
                           [('mpool', 3, 2, 'valid')]]),               # sc
            _inc4_a(384), _inc4_a(384), _inc4_a(384), _inc4_a(384),
            _IncCols(384, [                                            # ra
                [('mpool', 3, 2, 'valid')],
                [('conv', 384, 3, 3, 2, 'valid')],
                [('conv', 192, 1, 1), ('conv', 224, 3, 3),
                 ('conv', 256, 3, 3, 2, 'valid')]]),
            _inc4_b(1024), _inc4_b(1024), _inc4_b(1024), _inc4_b(1024),
            _inc4_b(1024), _inc4_b(1024), _inc4_b(1024),
            _IncCols(1024, [                                           # rb
                [('mpool', 3, 2, 'valid')],
                [('conv', 192, 1, 1), ('conv', 192, 3, 3, 2, 'valid')],
                [('conv', 256, 1, 1), ('conv', 256, 1, 7),
                 ('conv', 320, 7, 1), ('conv', 320, 3, 3, 2, 'valid')]]),
            _inc4_c(1536), _inc4_c(1536), _inc4_c(1536))
        self.head = nn.Sequential(nn.AdaptiveAvgPool2d(1), nn.Flatten(),
                                  nn.Dropout(0.2),
                                  nn.Linear(1536, num_classes))

    def forward(self, x):
        return self.head(self.net(x))


# --- CIFAR-10 family --------------------------------------------------------

class _CifarResBlock(nn.Module):
    """resnet_model.py:40-83 — identity shortcut, Plan-A (1x1 avgpool
    subsample + zero channel-padding) when widening; v2 = pre-activation,
    no final ReLU."""

    def __init__(self, cin, depth, stride, pre_act):
        super().__init__()
        self.pre_act = pre_act
        self.widen = cin != depth
        self.stride = stride
        self.pad = (depth - cin) // 2 if self.widen else 0
        if pre_act:
            self.bn0 = nn.BatchNorm2d(cin, momentum=0.1, eps=1e-5)
        self.conv1 = _cbr(cin, depth, 3, stride)
        self.conv2 = nn.Conv2d(depth, depth, 3, padding=1, bias=False)
        if not pre_act:
            self.bn2 = nn.BatchNorm2d(depth, momentum=0.1, eps=1e-5)

    def forward(self, x):
        if self.widen:
            sc = F.avg_pool2d(x, 1, self.stride)
            sc = F.pad(sc, (0, 0, 0, 0, self.pad, self.pad))
        else:
            sc = x
        if self.pre_act:
            h = self.conv2(self.conv1(F.relu(self.bn0(x))))
            return sc + h
        h = self.bn2(self.conv2(self.conv1(x)))
        return F.relu(sc + h)


class CifarResNet(nn.Module):
    """resnet_model.py:281-331 — 16/32/64 channels, n blocks per stage
    (depth = 6n+2), 32x32 input."""

    def __init__(self, n, num_classes=11, pre_act=False):
        super().__init__()
        self.pre_act = pre_act
        if pre_act:
            self.stem = _cbr(3, 16, 3)
            self.bn_f = nn.BatchNorm2d(64, momentum=0.1, eps=1e-5)
        else:
            self.stem = nn.Sequential(
                nn.Conv2d(3, 16, 3, padding=1, bias=False),
                nn.BatchNorm2d(16, momentum=0.1, eps=1e-5))
        blocks, cin = [], 16
        for depth in (16, 32, 64):
            for i in range(n):
                s = 2 if depth > 16 and i == 0 else 1
                blocks.append(_CifarResBlock(cin, depth, s, pre_act))
                cin = depth
        self.blocks = nn.Sequential(*blocks)
        self.fc = nn.Linear(64, num_classes)

    def forward(self, x):
        h = self.blocks(self.stem(x))
        if self.pre_act:
            h = F.relu(self.bn_f(h))
        return self.fc(h.mean(dim=(2, 3)))


class _DenseLayer(nn.Module):
    def __init__(self, cin, growth):
        super().__init__()
        self.bn = nn.BatchNorm2d(cin, momentum=0.1, eps=1e-5)
        self.conv = nn.Conv2d(cin, growth, 3, padding=1, bias=False)

    def forward(self, x):
        return torch.cat([x, self.conv(F.relu(self.bn(x)))], 1)


class CifarDenseNet(nn.Module):
    """densenet_model.py:28-103 — BN-ReLU-conv3x3 dense layers, 1x1
    transition (no compression) + 2x2 avgpool."""

    def __init__(self, layer_counts, growth, num_classes=11):
        super().__init__()
        layers = [nn.Conv2d(3, 16, 3, padding=1)]
        c = 16
        for bi, n in enumerate(layer_counts):
            for _ in range(n):
                layers.append(_DenseLayer(c, growth))
                c += growth
            if bi < len(layer_counts) - 1:
                layers += [nn.BatchNorm2d(c, momentum=0.1, eps=1e-5),
                           nn.ReLU(inplace=True),
                           nn.Conv2d(c, c, 1, bias=False),
                           nn.AvgPool2d(2, 2)]
        layers += [nn.BatchNorm2d(c, momentum=0.1, eps=1e-5),
                   nn.ReLU(inplace=True)]
        self.features = nn.Sequential(*layers)
        self.fc = nn.Linear(c, num_classes)

    def forward(self, x):
        return self.fc(self.features(x).mean(dim=(2, 3)))


class AlexNetCifar(nn.Module):
    """alexnet_model.py:52-77 — cifar10-tutorial layout with LRN
    (depth_radius 4 -> size 9; TF alpha is un-normalised -> x9)."""

    def __init__(self, num_classes=11):
        super().__init__()
        lrn = nn.LocalResponseNorm(9, alpha=0.001, beta=0.75, k=1.0)
        self.features = nn.Sequential(
            nn.Conv2d(3, 64, 5, padding=2), nn.ReLU(inplace=True),
            nn.MaxPool2d(3, 2, 1), lrn,
            nn.Conv2d(64, 64, 5, padding=2), nn.ReLU(inplace=True), lrn,
            nn.MaxPool2d(3, 2, 1))
        self.classifier = nn.Sequential(
            nn.Flatten(), nn.Linear(64 * 8 * 8, 384),
            nn.ReLU(inplace=True), nn.Linear(384, 192),
            nn.ReLU(inplace=True), nn.Linear(192, num_classes))

    def forward(self, x):
        return self.classifier(self.features(x))


class TrivialCifar(nn.Module):
    def __init__(self, num_classes=11):
        super().__init__()
        self.net = nn.Sequential(nn.Flatten(), nn.Linear(3 * 32 * 32, 1),
                                 nn.Linear(1, num_classes))

    def forward(self, x):
        return self.net(x)


# cifar10 registry (model_config.py:48-63); the main registry additions
# are appended from models/cnn.py.
MODEL_REGISTRY_CIFAR = {
    'alexnet': lambda nc: AlexNetCifar(nc),
    'trivial': lambda nc: TrivialCifar(nc),
    'resnet20': lambda nc: CifarResNet(3, nc),
    'resnet20_v2': lambda nc: CifarResNet(3, nc, pre_act=True),
    'resnet32': lambda nc: CifarResNet(5, nc),
    'resnet32_v2': lambda nc: CifarResNet(5, nc, pre_act=True),
    'resnet44': lambda nc: CifarResNet(7, nc),
    'resnet44_v2': lambda nc: CifarResNet(7, nc, pre_act=True),
    'resnet56': lambda nc: CifarResNet(9, nc),
    'resnet56_v2': lambda nc: CifarResNet(9, nc, pre_act=True),
    'resnet110': lambda nc: CifarResNet(18, nc),
    'resnet110_v2': lambda nc: CifarResNet(18, nc, pre_act=True),
    'densenet40_k12': lambda nc: CifarDenseNet((12, 12, 12), 12, nc),
    'densenet100_k12': lambda nc: CifarDenseNet((32, 32, 32), 12, nc),
    'densenet100_k24': lambda nc: CifarDenseNet((32, 32, 32), 24, nc),
}
