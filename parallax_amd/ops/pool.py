"""NHWC pooling on hand-written HIP kernels (SURVEY §2.4 "max/avg
pool"; reference convnet_builder.py mpool/apool).  Covers the ResNet
uses: MaxPool2d(3, stride=2, padding=1) after the stem and the global
average pool before the FC.  channels_last bf16 with C % 8 == 0 runs
the kernels; anything else falls back to torch."""

import torch
import torch.nn as nn

from parallax_amd import ops as _ops


class _MaxPool3x3s2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2d, N, H, W, C):
        y, arg = _ops.ext().maxpool3x3s2_fwd(x2d, N, H, W, C)
        ctx.save_for_backward(arg)
        ctx.dims = (N, H, W, C)
        return y

    @staticmethod
    def backward(ctx, dy):
        (arg,) = ctx.saved_tensors
        N, H, W, C = ctx.dims
        dx = _ops.ext().maxpool3x3s2_bwd(dy.contiguous(), arg, N, H, W, C)
        return dx, None, None, None, None


class ParallaxMaxPool3x3s2(nn.Module):
    """Drop-in nn.MaxPool2d(3, stride=2, padding=1) for NHWC bf16."""

    def _native_ok(self, x):
        return (x.is_cuda and x.dtype == torch.bfloat16
                and x.shape[1] % 8 == 0 and _ops.native_available())

    def forward(self, x):
        N, C, H, W = x.shape
        if not self._native_ok(x):
            return torch.nn.functional.max_pool2d(x, 3, 2, 1)
        x2d = (x.permute(0, 2, 3, 1).reshape(N * H * W, C).contiguous()
               .view(-1))
        y = _MaxPool3x3s2Fn.apply(x2d, N, H, W, C)
        oh, ow = (H + 1) // 2, (W + 1) // 2
        return y.view(N, oh, ow, C).permute(0, 3, 1, 2)


class _GAvgPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2d, N, HW, C):
        ctx.dims = (N, HW, C)
        return _ops.ext().gavgpool_fwd(x2d, N, HW, C)

    @staticmethod
    def backward(ctx, dy):
        N, HW, C = ctx.dims
        dx = _ops.ext().gavgpool_bwd(dy, N, HW, C)
        return dx, None, None, None


class ParallaxGlobalAvgPool(nn.Module):
    """nn.AdaptiveAvgPool2d(1) + flatten for NHWC bf16 (returns
    [N, C] fp32 — feeds the FC head at full precision)."""

    def forward(self, x):
        N, C, H, W = x.shape
        if not (x.is_cuda and x.dtype == torch.bfloat16 and C % 8 == 0
                and _ops.native_available()):
            return torch.nn.functional.adaptive_avg_pool2d(x, 1).flatten(1)
        x2d = (x.permute(0, 2, 3, 1).reshape(N * H * W, C).contiguous()
               .view(-1))
        return _GAvgPoolFn.apply(x2d, N, H * W, C)
