"""Bisect the native-conv1x1 ResNet segfault: run the pieces separately."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn

from parallax_amd.ops.conv import ParallaxConv1x1, conv1x1_nhwc

dev = 'cuda:0'
torch.manual_seed(0)


def t(tag, fn):
    try:
        fn()
        torch.cuda.synchronize()
        print('%s: OK' % tag, flush=True)
    except Exception as e:
        print('%s: EXC %r' % (tag, e), flush=True)


# 1. conv1x1 + BN at the exact first-bottleneck shape, batch 2
def case1():
    conv = ParallaxConv1x1(64, 64).to(dev)
    bn = nn.BatchNorm2d(64, momentum=0.001).to(dev)
    conv = conv.to(memory_format=torch.channels_last)
    bn = bn.to(memory_format=torch.channels_last)
    x = torch.randn(2, 64, 56, 56, device=dev).to(
        memory_format=torch.channels_last)
    with torch.autocast('cuda', dtype=torch.bfloat16):
        y = conv(x)
        print('  conv out', y.shape, y.dtype, y.stride(), flush=True)
        z = bn(y)
    z.float().sum().backward()


t('case1 conv64->64 + bn, batch2', case1)


# 2. same with batch 64
def case2():
    conv = ParallaxConv1x1(64, 64).to(dev)
    bn = nn.BatchNorm2d(64, momentum=0.001).to(dev)
    x = torch.randn(64, 64, 56, 56, device=dev).to(
        memory_format=torch.channels_last)
    with torch.autocast('cuda', dtype=torch.bfloat16):
        z = bn(conv(x))
    z.float().sum().backward()


t('case2 batch64', case2)


# 3. raw pxgemm at the batch-2 shape incl pad path
def case3():
    from parallax_amd import ops
    e = ops.ext()
    a = torch.randn(6272, 64, device=dev, dtype=torch.bfloat16)
    b = torch.randn(128, 64, device=dev, dtype=torch.bfloat16)
    c = e.pxgemm_bt(a, b)
    ref = a.float() @ b.float().t()
    print('  err', float((c.float() - ref).abs().max()), flush=True)


t('case3 raw gemm 6272x128x64', case3)


# 4. full swapped resnet fwd only, batch 2, no autocast
def case4():
    os.environ['PARALLAX_NATIVE_CONV1X1'] = '1'
    from parallax_amd.models.resnet import build_resnet_graph, \
        synthetic_batch
    g = build_resnet_graph(depth=50, batch_size=2, use_bf16=True)
    g.model.to(dev)
    x, y = synthetic_batch(2, device=dev)
    loss, _ = g.step_fn(g.model, (x, y))
    loss.backward()
    print('  loss', float(loss), flush=True)


t('case4 full swapped resnet batch2', case4)
print('ALL DONE', flush=True)
