"""NHWC BatchNorm2d (+ optional fused ReLU) on hand-written HIP kernels.

SURVEY §2.4 worklist entry "BatchNorm fwd/bwd (decay .999) + ReLU"
(reference convnet_builder.py:428-501).  Training-mode statistics,
normalize and backward run on the kernels in ops/hip/csrc/batchnorm.hip;
eval mode and CPU use the torch composition.  State-dict layout matches
nn.BatchNorm2d (weight/bias/running_mean/running_var/num_batches_tracked)
for checkpoint parity, and ``from_bn`` converts an existing module.

Also the workaround for the documented MIOpen segfault on tiny bf16
channels_last BN shapes (doc/design.md).
"""

import torch
import torch.nn as nn

from parallax_amd import ops as _ops


class _BNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2d, weight, bias, run_mean, run_var, momentum, eps,
                relu):
        y, mean, invstd = _ops.ext().batchnorm_fwd(
            x2d, weight, bias, run_mean, run_var, momentum, eps, relu)
        ctx.save_for_backward(x2d, mean, invstd, weight, bias)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x2d, mean, invstd, weight, bias = ctx.saved_tensors
        dx, dw, db = _ops.ext().batchnorm_bwd(dy, x2d, mean, invstd,
                                              weight, bias, ctx.relu)
        return (dx, dw.to(weight.dtype), db.to(bias.dtype), None, None,
                None, None, None)


class ParallaxBatchNorm2d(nn.Module):
    """Drop-in nn.BatchNorm2d for NHWC power-of-two channel counts, with
    ``relu=True`` fusing the activation into the normalize/backward."""

    def __init__(self, num_features, eps=1e-5, momentum=0.001, relu=False):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.relu = relu
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer('running_mean', torch.zeros(num_features))
        self.register_buffer('running_var', torch.ones(num_features))
        self.register_buffer('num_batches_tracked',
                             torch.tensor(0, dtype=torch.long))

    @classmethod
    def from_bn(cls, bn, relu=False):
        m = cls(bn.num_features, bn.eps, bn.momentum, relu)
        with torch.no_grad():
            m.weight.copy_(bn.weight)
            m.bias.copy_(bn.bias)
            m.running_mean.copy_(bn.running_mean)
            m.running_var.copy_(bn.running_var)
        return m

    def _native_ok(self, x):
        c = self.num_features
        return (x.is_cuda and _ops.native_available() and self.training
                and (c & (c - 1)) == 0 and 64 <= c <= 2048)

    def forward(self, x):
        N, C, H, W = x.shape
        if self._native_ok(x):
            self.num_batches_tracked += 1
            x2d = (x.permute(0, 2, 3, 1).reshape(N * H * W, C)
                   .contiguous())
            y = _BNFn.apply(x2d, self.weight, self.bias,
                            self.running_mean, self.running_var,
                            self.momentum, self.eps, self.relu)
            return y.view(N, H, W, C).permute(0, 3, 1, 2)
        y = torch.nn.functional.batch_norm(
            x, self.running_mean, self.running_var, self.weight,
            self.bias, self.training, self.momentum, self.eps)
        return torch.relu(y) if self.relu else y
