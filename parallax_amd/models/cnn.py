"""CNN benchmark model family — the tf_cnn_benchmarks equivalents.

Reference: examples/tf_cnn_benchmarks (models/*.py: ResNet-50/101/152
v1/v2, VGG-16/19, AlexNet, GoogLeNet, Inception-v3, Overfeat, Trivial;
registry in benchmark_cnn.py).  All dense models -> they exercise the
bucketed RCCL all-reduce engine (MPI mode / degenerate HYBRID).

``build_cnn_graph(model_name, ...)`` is the benchmark_cnn-style entry:
synthetic ImageNet, softmax-CE (fused HIP kernel), SGD/Momentum/RMSProp
choice (benchmark_cnn.py:805-818), NHWC (channels_last) + bf16 on GPU.
"""

import torch
import torch.nn as nn

from parallax_amd import optim
from parallax_amd.models.resnet import ResNet
from parallax_amd.ops.softmax_ce import softmax_cross_entropy
from parallax_amd.session import TrainGraph


def _conv_bn_relu(cin, cout, k, s=1, p=None):
    if p is None:
        p = k // 2
    return nn.Sequential(
        nn.Conv2d(cin, cout, k, stride=s, padding=p, bias=False),
        nn.BatchNorm2d(cout, momentum=0.001), nn.ReLU(inplace=True))


class VGG16(nn.Module):
    CFG = [64, 64, 'M', 128, 128, 'M', 256, 256, 256, 'M',
           512, 512, 512, 'M', 512, 512, 512, 'M']

    def __init__(self, num_classes=1001):
        super().__init__()
        cfg = self.CFG
        layers, cin = [], 3
        for v in cfg:
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


class VGG19(VGG16):
    CFG = [64, 64, 'M', 128, 128, 'M', 256, 256, 256, 256, 'M',
           512, 512, 512, 512, 'M', 512, 512, 512, 512, 'M']


class AlexNet(nn.Module):
    def __init__(self, num_classes=1001):
        super().__init__()
        self.features = nn.Sequential(
            nn.Conv2d(3, 64, 11, 4, 2), nn.ReLU(inplace=True),
            nn.MaxPool2d(3, 2),
            nn.Conv2d(64, 192, 5, padding=2), nn.ReLU(inplace=True),
            nn.MaxPool2d(3, 2),
            nn.Conv2d(192, 384, 3, padding=1), nn.ReLU(inplace=True),
            nn.Conv2d(384, 256, 3, padding=1), nn.ReLU(inplace=True),
            nn.Conv2d(256, 256, 3, padding=1), nn.ReLU(inplace=True),
            nn.MaxPool2d(3, 2))
        self.classifier = nn.Sequential(
            nn.Flatten(), nn.Dropout(0.5), nn.Linear(256 * 6 * 6, 4096),
            nn.ReLU(inplace=True), nn.Dropout(0.5), nn.Linear(4096, 4096),
            nn.ReLU(inplace=True), nn.Linear(4096, num_classes))

    def forward(self, x):
        return self.classifier(self.features(x))


class Overfeat(nn.Module):
    def __init__(self, num_classes=1001):
        super().__init__()
        self.features = nn.Sequential(
            nn.Conv2d(3, 96, 11, 4), nn.ReLU(inplace=True),
            nn.MaxPool2d(2, 2),
            nn.Conv2d(96, 256, 5), nn.ReLU(inplace=True),
            nn.MaxPool2d(2, 2),
            nn.Conv2d(256, 512, 3, padding=1), nn.ReLU(inplace=True),
            nn.Conv2d(512, 1024, 3, padding=1), nn.ReLU(inplace=True),
            nn.Conv2d(1024, 1024, 3, padding=1), nn.ReLU(inplace=True),
            nn.MaxPool2d(2, 2))
        self.classifier = nn.Sequential(
            nn.Flatten(), nn.Linear(1024 * 6 * 6, 3072),
            nn.ReLU(inplace=True), nn.Linear(3072, 4096),
            nn.ReLU(inplace=True), nn.Linear(4096, num_classes))

    def forward(self, x):
        return self.classifier(self.features(x))


class _Inception(nn.Module):
    def __init__(self, cin, c1, c3r, c3, c5r, c5, pp):
        super().__init__()
        self.b1 = _conv_bn_relu(cin, c1, 1)
        self.b3 = nn.Sequential(_conv_bn_relu(cin, c3r, 1),
                                _conv_bn_relu(c3r, c3, 3))
        self.b5 = nn.Sequential(_conv_bn_relu(cin, c5r, 1),
                                _conv_bn_relu(c5r, c5, 5))
        self.bp = nn.Sequential(nn.MaxPool2d(3, 1, 1),
                                _conv_bn_relu(cin, pp, 1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b3(x), self.b5(x), self.bp(x)],
                         dim=1)


class GoogLeNet(nn.Module):
    def __init__(self, num_classes=1001):
        super().__init__()
        self.stem = nn.Sequential(
            _conv_bn_relu(3, 64, 7, 2), nn.MaxPool2d(3, 2, 1),
            _conv_bn_relu(64, 64, 1), _conv_bn_relu(64, 192, 3),
            nn.MaxPool2d(3, 2, 1))
        self.blocks = nn.Sequential(
            _Inception(192, 64, 96, 128, 16, 32, 32),
            _Inception(256, 128, 128, 192, 32, 96, 64),
            nn.MaxPool2d(3, 2, 1),
            _Inception(480, 192, 96, 208, 16, 48, 64),
            _Inception(512, 160, 112, 224, 24, 64, 64),
            _Inception(512, 128, 128, 256, 24, 64, 64),
            _Inception(512, 112, 144, 288, 32, 64, 64),
            _Inception(528, 256, 160, 320, 32, 128, 128),
            nn.MaxPool2d(3, 2, 1),
            _Inception(832, 256, 160, 320, 32, 128, 128),
            _Inception(832, 384, 192, 384, 48, 128, 128))
        self.head = nn.Sequential(nn.AdaptiveAvgPool2d(1), nn.Flatten(),
                                  nn.Dropout(0.4),
                                  nn.Linear(1024, num_classes))

    def forward(self, x):
        return self.head(self.blocks(self.stem(x)))


class TrivialModel(nn.Module):
    """tf_cnn_benchmarks 'trivial' (pipeline-overhead probe)."""

    def __init__(self, num_classes=1001, image_size=224):
        super().__init__()
        self.net = nn.Sequential(nn.Flatten(),
                                 nn.Linear(3 * image_size * image_size, 1),
                                 nn.Linear(1, num_classes))

    def forward(self, x):
        return self.net(x)


MODEL_REGISTRY = {
    'resnet50': lambda nc: ResNet(50, nc),
    'resnet101': lambda nc: ResNet(101, nc),
    'resnet152': lambda nc: ResNet(152, nc),
    'resnet50_v2': lambda nc: ResNet(50, nc, version=2),
    'resnet101_v2': lambda nc: ResNet(101, nc, version=2),
    'resnet152_v2': lambda nc: ResNet(152, nc, version=2),
    'vgg16': lambda nc: VGG16(nc),
    'vgg19': lambda nc: VGG19(nc),
    'alexnet': lambda nc: AlexNet(nc),
    'overfeat': lambda nc: Overfeat(nc),
    'googlenet': lambda nc: GoogLeNet(nc),
    'trivial': lambda nc: TrivialModel(nc),
}


def topk_accuracy(logits, labels, ks=(1, 5)):
    """Fraction of rows whose label is in the top-k logits
    (benchmark_cnn.py top_1/top_5 accuracy fetches)."""
    out = {}
    kmax = min(max(ks), logits.shape[-1])
    top = logits.detach().topk(kmax, dim=-1).indices
    for k in ks:
        hit = (top[:, :min(k, kmax)] == labels[:, None]).any(dim=1)
        out['top_%d' % k] = hit.float().mean()
    return out


def build_cnn_graph(model_name='resnet50', num_classes=1001, batch_size=64,
                    optimizer_name='momentum', lr=0.1, momentum=0.9,
                    weight_decay=1e-4, use_bf16=True, channels_last=True,
                    pure_bf16=False, dataset='imagenet',
                    with_accuracy=False):
    """pure_bf16: conv/linear weights live in bf16 (no per-step autocast
    weight casts; fp32 update precision via the optimizers' fused master
    weights); BatchNorm modules stay fp32 (weights + running stats — the
    mixed bf16-input/fp32-param BN is the library's standard
    mixed-precision path)."""
    model = get_model(model_name, num_classes, dataset)
    if channels_last:
        model = model.to(memory_format=torch.channels_last)
    if pure_bf16:
        model = model.to(torch.bfloat16)
        for m in model.modules():
            if isinstance(m, nn.modules.batchnorm._BatchNorm):
                m.float()
    if optimizer_name == 'momentum':
        opt = optim.SGD(model.parameters(), lr=lr, momentum=momentum,
                        weight_decay=weight_decay)
    elif optimizer_name == 'sgd':
        opt = optim.SGD(model.parameters(), lr=lr,
                        weight_decay=weight_decay)
    elif optimizer_name == 'rmsprop':
        opt = optim.RMSProp(model.parameters(), lr=lr, decay=0.9,
                            momentum=momentum, eps=1.0)
    else:
        raise ValueError('optimizer must be momentum|sgd|rmsprop '
                         '(benchmark_cnn.py:805-818)')

    def step_fn(m, batch):
        x, y = batch
        if channels_last and x.dim() == 4:
            x = x.to(memory_format=torch.channels_last)
        if pure_bf16:
            logits = m(x.to(torch.bfloat16))
            loss = softmax_cross_entropy(logits.float(), y)
        elif use_bf16 and x.is_cuda:
            with torch.autocast('cuda', dtype=torch.bfloat16):
                logits = m(x)
            loss = softmax_cross_entropy(logits.float(), y)
        else:
            logits = m(x)
            loss = softmax_cross_entropy(logits, y)
        fetches = topk_accuracy(logits.float(), y) if with_accuracy else {}
        return loss, fetches

    return TrainGraph(model, opt, step_fn)


# --- Inception v3 (reference models/inception_model.py) ---------------------

class _IncA(nn.Module):
    def __init__(self, cin, pool_ch):
        super().__init__()
        self.b1 = _conv_bn_relu(cin, 64, 1)
        self.b5 = nn.Sequential(_conv_bn_relu(cin, 48, 1),
                                _conv_bn_relu(48, 64, 5))
        self.b3 = nn.Sequential(_conv_bn_relu(cin, 64, 1),
                                _conv_bn_relu(64, 96, 3),
                                _conv_bn_relu(96, 96, 3))
        self.bp = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                _conv_bn_relu(cin, pool_ch, 1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b5(x), self.b3(x), self.bp(x)],
                         1)


class _IncB(nn.Module):  # grid reduction 35->17
    def __init__(self, cin):
        super().__init__()
        self.b3 = _conv_bn_relu(cin, 384, 3, s=2, p=0)
        self.bd = nn.Sequential(_conv_bn_relu(cin, 64, 1),
                                _conv_bn_relu(64, 96, 3),
                                _conv_bn_relu(96, 96, 3, s=2, p=0))
        self.mp = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b3(x), self.bd(x), self.mp(x)], 1)


class _IncC(nn.Module):  # factorized 7x7
    def __init__(self, cin, ch7):
        super().__init__()
        def c7(cin_, cout, horiz):
            k, p = ((1, 7), (0, 3)) if horiz else ((7, 1), (3, 0))
            return nn.Sequential(nn.Conv2d(cin_, cout, k, padding=p,
                                           bias=False),
                                 nn.BatchNorm2d(cout, momentum=0.001),
                                 nn.ReLU(inplace=True))
        self.b1 = _conv_bn_relu(cin, 192, 1)
        self.b7 = nn.Sequential(_conv_bn_relu(cin, ch7, 1),
                                c7(ch7, ch7, True), c7(ch7, 192, False))
        self.b77 = nn.Sequential(_conv_bn_relu(cin, ch7, 1),
                                 c7(ch7, ch7, False), c7(ch7, ch7, True),
                                 c7(ch7, ch7, False), c7(ch7, 192, True))
        self.bp = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                _conv_bn_relu(cin, 192, 1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b7(x), self.b77(x),
                          self.bp(x)], 1)


class _IncD(nn.Module):  # grid reduction 17->8
    def __init__(self, cin):
        super().__init__()
        self.b3 = nn.Sequential(_conv_bn_relu(cin, 192, 1),
                                _conv_bn_relu(192, 320, 3, s=2, p=0))
        self.b7 = nn.Sequential(
            _conv_bn_relu(cin, 192, 1),
            nn.Conv2d(192, 192, (1, 7), padding=(0, 3), bias=False),
            nn.BatchNorm2d(192, momentum=0.001), nn.ReLU(inplace=True),
            nn.Conv2d(192, 192, (7, 1), padding=(3, 0), bias=False),
            nn.BatchNorm2d(192, momentum=0.001), nn.ReLU(inplace=True),
            _conv_bn_relu(192, 192, 3, s=2, p=0))
        self.mp = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b3(x), self.b7(x), self.mp(x)], 1)


class _IncE(nn.Module):  # expanded 8x8 blocks
    def __init__(self, cin):
        super().__init__()
        self.b1 = _conv_bn_relu(cin, 320, 1)
        self.b3_stem = _conv_bn_relu(cin, 384, 1)
        self.b3_a = nn.Sequential(
            nn.Conv2d(384, 384, (1, 3), padding=(0, 1), bias=False),
            nn.BatchNorm2d(384, momentum=0.001), nn.ReLU(inplace=True))
        self.b3_b = nn.Sequential(
            nn.Conv2d(384, 384, (3, 1), padding=(1, 0), bias=False),
            nn.BatchNorm2d(384, momentum=0.001), nn.ReLU(inplace=True))
        self.bd_stem = nn.Sequential(_conv_bn_relu(cin, 448, 1),
                                     _conv_bn_relu(448, 384, 3))
        self.bd_a = nn.Sequential(
            nn.Conv2d(384, 384, (1, 3), padding=(0, 1), bias=False),
            nn.BatchNorm2d(384, momentum=0.001), nn.ReLU(inplace=True))
        self.bd_b = nn.Sequential(
            nn.Conv2d(384, 384, (3, 1), padding=(1, 0), bias=False),
            nn.BatchNorm2d(384, momentum=0.001), nn.ReLU(inplace=True))
        self.bp = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                _conv_bn_relu(cin, 192, 1))

    def forward(self, x):
        s3 = self.b3_stem(x)
        sd = self.bd_stem(x)
        return torch.cat([self.b1(x), self.b3_a(s3), self.b3_b(s3),
                          self.bd_a(sd), self.bd_b(sd), self.bp(x)], 1)


class InceptionV3(nn.Module):
    """Inception-v3 (299x299 input)."""

    def __init__(self, num_classes=1001):
        super().__init__()
        self.stem = nn.Sequential(
            _conv_bn_relu(3, 32, 3, s=2, p=0),
            _conv_bn_relu(32, 32, 3, p=0),
            _conv_bn_relu(32, 64, 3),
            nn.MaxPool2d(3, 2),
            _conv_bn_relu(64, 80, 1),
            _conv_bn_relu(80, 192, 3, p=0),
            nn.MaxPool2d(3, 2))
        self.blocks = nn.Sequential(
            _IncA(192, 32), _IncA(256, 64), _IncA(288, 64),
            _IncB(288),
            _IncC(768, 128), _IncC(768, 160), _IncC(768, 160),
            _IncC(768, 192),
            _IncD(768),
            _IncE(1280), _IncE(2048))
        self.head = nn.Sequential(nn.AdaptiveAvgPool2d(1), nn.Flatten(),
                                  nn.Dropout(0.2),
                                  nn.Linear(2048, num_classes))

    def forward(self, x):
        return self.head(self.blocks(self.stem(x)))


MODEL_REGISTRY['inception3'] = lambda nc: InceptionV3(nc)

# vgg11 / lenet / inception4 + the cifar10 registry
# (models/model_config.py:29-63) live in cnn_extra.py
from parallax_amd.models.cnn_extra import (  # noqa: E402
    VGG11, LeNet5, InceptionV4, MODEL_REGISTRY_CIFAR)

MODEL_REGISTRY['vgg11'] = lambda nc: VGG11(nc)
MODEL_REGISTRY['lenet'] = lambda nc: LeNet5(nc)
MODEL_REGISTRY['inception4'] = lambda nc: InceptionV4(nc)

# per-model input resolution (reference models/*.py image_size)
MODEL_IMAGE_SIZE = {'overfeat': 231, 'inception3': 299, 'inception4': 299,
                    'lenet': 28}


def image_size_for(model_name, dataset='imagenet'):
    if dataset == 'cifar10':
        return 32
    return MODEL_IMAGE_SIZE.get(model_name, 224)


def get_model(model_name, num_classes, dataset='imagenet'):
    """model_config.get_model_config equivalent: dataset-keyed lookup
    ('alexnet'/'trivial' exist in both registries with different nets)."""
    reg = MODEL_REGISTRY_CIFAR if dataset == 'cifar10' else MODEL_REGISTRY
    if model_name not in reg:
        raise ValueError('unknown model %r for dataset %r'
                         % (model_name, dataset))
    return reg[model_name](num_classes)
