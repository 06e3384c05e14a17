"""NHWC convolutions on the hand-written MFMA GEMM family
(SURVEY §2.4 conv worklist).

1x1: fwd + dgrad on the tile-dispatched GEMMs (_best_gemm), wgrad on
MIOpen wrw (hipBLASLt picks a ~5x-slower TN split-K schedule there).
3x3: complete implicit-GEMM path — fwd and stride-1 dgrad on the
tall-tile kernel over a host-padded halo (dgrad = same kernel on
pad(dy) with the rotated/transposed weight), wgrad on px_wgrad3x3
(LDS-transposed staging + split-M fp32 atomics).

Both stay OPT-IN (PARALLAX_NATIVE_CONV1X1/3X3=1): numerics-tested and
isolated-faster on the big GEMM shapes, but the exhaustive-find-tuned
MIOpen conv suite still wins end-to-end ResNet (honest A/Bs in
doc/design.md).
"""

import torch
import torch.nn as nn

from parallax_amd import ops as _ops


def _best_gemm(e, a, b):
    """Tile-family dispatch: the 256x64 tall tile for M-huge/N-small
    (fills the chip where the wide tiles waste columns, measured
    profiles/tall_ab_r2.log), 256^2 8-phase otherwise."""
    M, N = a.shape[0], b.shape[0]
    if (hasattr(e, 'px_tall_bt') and M >= 8 * N
            and (M // 256) * ((N + 63) // 64) >= 256):
        return e.px_tall_bt(a, b)
    if hasattr(e, 'pxgemm8_bt') and M >= 512:
        return e.pxgemm8_bt(a, b)
    return e.pxgemm_bt(a, b)


class _Conv1x1Fn(torch.autograd.Function):
    """fwd + dgrad on the 8-phase MFMA GEMM; wgrad through MIOpen's wrw
    igemm (hipBLASLt's TN split-K GEMM on this shape picks a ~5x-slower
    schedule — gpurun_out/pc11 profile)."""

    @staticmethod
    def forward(ctx, x, x2d, weight4):
        ctx.save_for_backward(x, weight4)
        e = _ops.ext()
        w2d = weight4.reshape(weight4.shape[0], weight4.shape[1])
        x2d = x2d.contiguous()
        w = w2d.contiguous()
        return _best_gemm(e, x2d, w)

    @staticmethod
    def backward(ctx, dy2d):
        x, weight4 = ctx.saved_tensors
        e = _ops.ext()
        dy2d = dy2d.contiguous()
        w2d = weight4.reshape(weight4.shape[0], weight4.shape[1])
        # dgrad on the MFMA kernel: dy @ W == dy @ (W^T)^T
        if (dy2d.shape[1] % 64 == 0 and hasattr(e, 'pxgemm8_bt')
                and dy2d.shape[0] >= 512):
            dx2d = _best_gemm(e, dy2d, w2d.t().contiguous())
        else:
            dx2d = dy2d @ w2d
        N, C, H, W = x.shape
        cout = weight4.shape[0]
        dy4 = (dy2d.view(N, H, W, cout).permute(0, 3, 1, 2))
        _, dw, _ = torch.ops.aten.convolution_backward(
            dy4, x, weight4.to(x.dtype), None, [1, 1], [0, 0], [1, 1],
            False, [0, 0], 1, [False, True, False])
        return None, dx2d, dw.to(weight4.dtype)


def conv1x1_nhwc(x, weight):
    """x: [N, C, H, W] channels_last bf16; weight: [Cout, Cin, 1, 1].
    Returns [N, Cout, H, W] channels_last."""
    N, C, H, W = x.shape
    cout = weight.shape[0]
    x2d = x.permute(0, 2, 3, 1).reshape(N * H * W, C)  # free in NHWC
    w2d = weight.reshape(cout, C)
    if x.is_cuda and _ops.native_available() \
            and x.dtype == torch.bfloat16 and C % 64 == 0:
        out = _Conv1x1Fn.apply(x, x2d, weight.to(x.dtype))
    else:
        out = x2d @ w2d.to(x2d.dtype).t()
    return out.view(N, H, W, cout).permute(0, 3, 1, 2)


class ParallaxConv1x1(nn.Module):
    """Drop-in for nn.Conv2d(k=1, stride=1, bias=False) on the MFMA GEMM."""

    def __init__(self, cin, cout):
        super().__init__()
        w = torch.empty(cout, cin, 1, 1)
        nn.init.kaiming_normal_(w, mode='fan_out', nonlinearity='relu')
        self.weight = nn.Parameter(w)
        self.in_channels = cin
        self.out_channels = cout

    @classmethod
    def from_conv(cls, conv):
        assert conv.kernel_size == (1, 1) and conv.stride == (1, 1) \
            and conv.bias is None
        m = cls(conv.in_channels, conv.out_channels)
        with torch.no_grad():
            m.weight.copy_(conv.weight)
        return m

    def forward(self, x):
        return conv1x1_nhwc(x, self.weight)


def swap_conv1x1(module):
    """Recursively replace eligible nn.Conv2d(1x1, stride 1, no bias)
    children with ParallaxConv1x1 (PARALLAX_NATIVE_CONV1X1=1 path)."""
    for name, child in module.named_children():
        if (isinstance(child, nn.Conv2d) and child.kernel_size == (1, 1)
                and child.stride == (1, 1) and child.bias is None
                and child.groups == 1 and child.dilation == (1, 1)):
            setattr(module, name, ParallaxConv1x1.from_conv(child))
        else:
            swap_conv1x1(child)
    return module


def _tall_grid(M, cout):
    return (M // 256) * ((cout + 63) // 64)


def _conv3x3_gemm(xp, w2d, N, oh, ow, C, stride, HP, WP):
    """Pick the tall-tile conv kernel when its grid can occupy the chip
    (>= 96 blocks at 2 blocks/CU), the 128^2 kernel otherwise."""
    e = _ops.ext()
    M = N * oh * ow
    cout = w2d.shape[0]
    if (hasattr(e, 'px_tall_conv3x3') and _tall_grid(M, cout) >= 96):
        return e.px_tall_conv3x3(xp, w2d, N, oh, ow, C, stride, HP, WP)
    return e.pxconv3x3_fwd(xp, w2d, N, oh, ow, C, stride, HP, WP)


class _Conv3x3Fn(torch.autograd.Function):
    """Forward: implicit-GEMM MFMA kernel over a host-padded NHWC input
    (tall 256x64 tile when the grid fills the chip).  Backward: stride-1
    dgrad is the SAME implicit GEMM over pad(dy) with the
    rotated/transposed weight; wgrad (and stride-2 dgrad) go through
    MIOpen (aten.convolution_backward with a masked output)."""

    @staticmethod
    def forward(ctx, x, weight, stride):
        ctx.save_for_backward(x, weight)
        ctx.stride = stride
        N, C, H, W = x.shape
        cout = weight.shape[0]
        oh = (H + 2 - 3) // stride + 1
        ow = (W + 2 - 3) // stride + 1
        # halo pad once per call (glds cannot predicate edge reads)
        x_nhwc = x.permute(0, 2, 3, 1)            # view (channels_last)
        xp = torch.nn.functional.pad(x_nhwc, (0, 0, 1, 1, 1, 1))
        w2d = (weight.permute(0, 2, 3, 1).reshape(cout, 9 * C)
               .to(x.dtype).contiguous())
        out = _conv3x3_gemm(xp.contiguous(), w2d, N, oh, ow, C,
                            stride, H + 2, W + 2)
        return out.view(N, oh, ow, cout).permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        s = ctx.stride
        N, C, H, W = x.shape
        cout = weight.shape[0]
        native_dgrad = (s == 1 and cout >= 64
                        and (cout & (cout - 1)) == 0
                        and _tall_grid(N * H * W, C) >= 96
                        and ctx.needs_input_grad[0])
        if native_dgrad:
            # dx = conv3x3(pad(dy), Wrot) with
            # Wrot[ci][dy][dx][co] = w[co][2-dy][2-dx][ci]
            dyc = dy.to(x.dtype)
            dy_nhwc = dyc.permute(0, 2, 3, 1)
            dyp = torch.nn.functional.pad(dy_nhwc, (0, 0, 1, 1, 1, 1))
            wrot = (weight.to(x.dtype).flip(2, 3)
                    .permute(1, 2, 3, 0).reshape(C, 9 * cout)
                    .contiguous())
            dx2 = _conv3x3_gemm(dyp.contiguous(), wrot, N, H, W, cout,
                                1, H + 2, W + 2)
            dx = (dx2.view(N, H, W, C).permute(0, 3, 1, 2))
            e = _ops.ext()
            if hasattr(e, 'px_wgrad3x3') and cout % 8 == 0:
                # native wgrad: LDS-transposed implicit GEMM over the
                # re-padded input; fp32 accumulate, one cast at the end
                oh, ow = H, W  # stride-1 branch: output size == input
                xp = torch.nn.functional.pad(
                    x.permute(0, 2, 3, 1), (0, 0, 1, 1, 1, 1))
                dy2 = (dy_nhwc.reshape(N * H * W, cout).contiguous())
                ws = e.px_wgrad3x3(dy2, xp.contiguous(), N, oh, ow, C,
                                   1, H + 2, W + 2)
                dw = (ws.view(cout, 3, 3, C).permute(0, 3, 1, 2)
                      .to(weight.dtype))
            else:
                _, dw, _ = torch.ops.aten.convolution_backward(
                    dyc, x, weight.to(x.dtype), None, [s, s], [1, 1],
                    [1, 1], False, [0, 0], 1, [False, True, False])
        else:
            dx, dw, _ = torch.ops.aten.convolution_backward(
                dy.to(x.dtype), x, weight.to(x.dtype), None, [s, s],
                [1, 1], [1, 1], False, [0, 0], 1,
                [ctx.needs_input_grad[0], True, False])
        return dx, dw.to(weight.dtype), None


def conv3x3_nhwc(x, weight, stride=1):
    """3x3 pad-1 NHWC convolution; forward on the hand-written MFMA
    implicit-GEMM kernel when eligible (bf16, Cin power-of-two >= 64)."""
    C = x.shape[1]
    N, _, H, W = x.shape
    oh = (H + 2 - 3) // stride + 1
    ow = (W + 2 - 3) // stride + 1
    if (x.is_cuda and _ops.native_available()
            and x.dtype == torch.bfloat16 and C >= 64
            and (C & (C - 1)) == 0
            and _tall_grid(N * oh * ow, weight.shape[0]) >= 96):
        return _Conv3x3Fn.apply(x, weight, stride)
    return torch.nn.functional.conv2d(x, weight.to(x.dtype),
                                      stride=stride, padding=1)


class ParallaxConv3x3(nn.Module):
    """Drop-in for nn.Conv2d(k=3, padding=1, bias=False)."""

    def __init__(self, cin, cout, stride=1):
        super().__init__()
        w = torch.empty(cout, cin, 3, 3)
        nn.init.kaiming_normal_(w, mode='fan_out', nonlinearity='relu')
        self.weight = nn.Parameter(w)
        self.stride = stride
        self.in_channels = cin
        self.out_channels = cout

    @classmethod
    def from_conv(cls, conv):
        assert conv.kernel_size == (3, 3) and conv.padding == (1, 1) \
            and conv.bias is None and conv.stride[0] == conv.stride[1]
        m = cls(conv.in_channels, conv.out_channels, conv.stride[0])
        with torch.no_grad():
            m.weight.copy_(conv.weight)
        return m

    def forward(self, x):
        return conv3x3_nhwc(x, self.weight, self.stride)


def swap_conv3x3(module):
    """Replace eligible nn.Conv2d(3x3, pad 1, no bias, power-of-two
    Cin >= 64) children with ParallaxConv3x3
    (PARALLAX_NATIVE_CONV3X3=1 path)."""
    for name, child in module.named_children():
        if (isinstance(child, nn.Conv2d) and child.kernel_size == (3, 3)
                and child.padding == (1, 1) and child.bias is None
                and child.groups == 1 and child.dilation == (1, 1)
                and child.stride[0] == child.stride[1] <= 2
                and child.in_channels >= 64
                and (child.in_channels & (child.in_channels - 1)) == 0):
            setattr(module, name, ParallaxConv3x3.from_conv(child))
        else:
            swap_conv3x3(child)
    return module
