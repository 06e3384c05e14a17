"""Second bisect: padded-M conv+bn shapes and per-layer swapped resnet."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn

from parallax_amd.ops.conv import ParallaxConv1x1

dev = 'cuda:0'
torch.manual_seed(0)


def t(tag, fn):
    try:
        fn()
        torch.cuda.synchronize()
        print('%s: OK' % tag, flush=True)
    except Exception as e:
        print('%s: EXC %r' % (tag, e), flush=True)


def conv_bn(cin, cout, hw, batch=2):
    def fn():
        conv = ParallaxConv1x1(cin, cout).to(dev).to(
            memory_format=torch.channels_last)
        bn = nn.BatchNorm2d(cout, momentum=0.001).to(dev)
        x = torch.randn(batch, cin, hw, hw, device=dev).to(
            memory_format=torch.channels_last)
        with torch.autocast('cuda', dtype=torch.bfloat16):
            z = bn(conv(x))
        z.float().sum().backward()
    return fn


for cin, cout, hw in ((64, 64, 56), (64, 256, 56), (256, 64, 56),
                      (256, 128, 28), (128, 512, 28), (512, 128, 28),
                      (512, 256, 14), (256, 1024, 14), (1024, 256, 14),
                      (1024, 512, 7), (512, 2048, 7), (2048, 512, 7)):
    t('conv_bn %d->%d @%d (M=%d)' % (cin, cout, hw, 2 * hw * hw),
      conv_bn(cin, cout, hw))

# per-layer swapped resnet
os.environ['PARALLAX_NATIVE_CONV1X1'] = '1'
from parallax_amd.models.resnet import ResNet
from parallax_amd.ops.conv import swap_conv1x1

m = ResNet(50, 1001)
swap_conv1x1(m)
m = m.to(memory_format=torch.channels_last).to(dev)
x = torch.randn(2, 3, 224, 224, device=dev).to(
    memory_format=torch.channels_last)
with torch.autocast('cuda', dtype=torch.bfloat16):
    h = m.maxpool(m.relu(m.bn1(m.conv1(x))))
torch.cuda.synchronize()
print('stem OK', h.shape, flush=True)
for name in ('layer1', 'layer2', 'layer3', 'layer4'):
    def fn(layer=getattr(m, name)):
        global h
        with torch.autocast('cuda', dtype=torch.bfloat16):
            hh = layer(h)
        torch.cuda.synchronize()
        return hh
    try:
        h = fn()
        print('%s: OK %s' % (name, tuple(h.shape)), flush=True)
    except Exception as e:
        print('%s: EXC %r' % (name, e), flush=True)
        break
print('ALL DONE', flush=True)
