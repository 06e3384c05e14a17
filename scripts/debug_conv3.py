"""Third bisect: is it our GEMM-conv, plain conv, or MIOpen BN?"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn

dev = 'cuda:0'
torch.manual_seed(0)


def t(tag, fn):
    try:
        fn()
        torch.cuda.synchronize()
        print('%s: OK' % tag, flush=True)
    except Exception as e:
        print('%s: EXC %r' % (tag, e), flush=True)


def bn_alone(c, hw, batch=2):
    def fn():
        bn = nn.BatchNorm2d(c, momentum=0.001).to(dev)
        x = (torch.randn(batch, c, hw, hw, device=dev, dtype=torch.bfloat16)
             .to(memory_format=torch.channels_last).requires_grad_(True))
        z = bn(x)
        z.float().sum().backward()
    return fn


def plain_conv_bn(cin, cout, hw, batch=2):
    def fn():
        conv = nn.Conv2d(cin, cout, 1, bias=False).to(dev).to(
            memory_format=torch.channels_last)
        bn = nn.BatchNorm2d(cout, momentum=0.001).to(dev)
        x = torch.randn(batch, cin, hw, hw, device=dev).to(
            memory_format=torch.channels_last)
        with torch.autocast('cuda', dtype=torch.bfloat16):
            z = bn(conv(x))
        z.float().sum().backward()
    return fn


def our_conv_alone(cin, cout, hw, batch=2):
    def fn():
        from parallax_amd.ops.conv import ParallaxConv1x1
        conv = ParallaxConv1x1(cin, cout).to(dev).to(
            memory_format=torch.channels_last)
        x = torch.randn(batch, cin, hw, hw, device=dev).to(
            memory_format=torch.channels_last)
        with torch.autocast('cuda', dtype=torch.bfloat16):
            z = conv(x)
        z.float().sum().backward()
    return fn


t('bn alone 512@7 b2', bn_alone(512, 7))
t('bn alone 2048@7 b2', bn_alone(2048, 7))
t('our conv alone 1024->512 @7 b2', our_conv_alone(1024, 512, 7))
t('our conv alone 512->2048 @7 b2', our_conv_alone(512, 2048, 7))
t('plain conv_bn 1024->512 @7 b2', plain_conv_bn(1024, 512, 7))
from parallax_amd.ops.conv import ParallaxConv1x1


def our_conv_bn(cin, cout, hw, batch=2):
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


t('our conv_bn 1024->512 @7 b2 (repeat)', our_conv_bn(1024, 512, 7))
print('ALL DONE', flush=True)
