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
    """Fused BN(+residual add)(+ReLU).  The saved OUTPUT y provides the
    ReLU mask in backward (with the add fused the mask depends on the
    post-add value); the identity branch's gradient dy' = dy * (y > 0)
    comes out of the same dx kernel pass."""

    @staticmethod
    def forward(ctx, x2d, weight, bias, run_mean, run_var, momentum, eps,
                relu, addend):
        y, mean, invstd = _ops.ext().batchnorm_fwd(
            x2d, weight, bias, run_mean, run_var, momentum, eps, relu,
            addend)
        ctx.save_for_backward(x2d, y, mean, invstd, weight)
        ctx.relu = relu
        ctx.has_add = addend is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x2d, y, mean, invstd, weight = ctx.saved_tensors
        out = _ops.ext().batchnorm_bwd(dy, x2d, y, mean, invstd, weight,
                                       ctx.relu, ctx.has_add)
        dx, dw, db = out[0], out[1], out[2]
        dxid = out[3] if ctx.has_add else None
        return (dx, dw.to(weight.dtype), db.to(weight.dtype), None, None,
                None, None, None, dxid)


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
        # forward() counts host-side (one CUDA add kernel per BN per
        # step otherwise — 53 launches/step in ResNet-50); the pending
        # count is flushed into the buffer at state_dict time
        self._nbt_pending = 0

    @classmethod
    def from_bn(cls, bn, relu=False):
        m = cls(bn.num_features, bn.eps, bn.momentum, relu)
        with torch.no_grad():
            m.weight.copy_(bn.weight)
            m.bias.copy_(bn.bias)
            m.running_mean.copy_(bn.running_mean)
            m.running_var.copy_(bn.running_var)
            if getattr(bn, 'num_batches_tracked', None) is not None:
                m.num_batches_tracked.copy_(bn.num_batches_tracked)
        return m

    def _native_ok(self, x):
        c = self.num_features
        return (x.is_cuda and _ops.native_available() and self.training
                and x.dtype in (torch.bfloat16, torch.float16)
                and c % 8 == 0 and c <= 2048)

    def _flush_nbt(self):
        if self._nbt_pending:
            self.num_batches_tracked += self._nbt_pending
            self._nbt_pending = 0

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        self._flush_nbt()
        super()._save_to_state_dict(destination, prefix, keep_vars)

    def _load_from_state_dict(self, *args, **kw):
        self._nbt_pending = 0
        super()._load_from_state_dict(*args, **kw)

    def forward(self, x, addend=None):
        """addend: optional residual tensor fused as
        y = [relu](bn(x) + addend) — the bottleneck's bn3 + shortcut."""
        N, C, H, W = x.shape
        if self._native_ok(x):
            self._nbt_pending += 1
            x2d = (x.permute(0, 2, 3, 1).reshape(N * H * W, C)
                   .contiguous())
            a2d = None
            if addend is not None:
                a2d = (addend.permute(0, 2, 3, 1).reshape(N * H * W, C)
                       .to(x.dtype).contiguous())
            y = _BNFn.apply(x2d, self.weight, self.bias,
                            self.running_mean, self.running_var,
                            self.momentum, self.eps, self.relu, a2d)
            return y.view(N, H, W, C).permute(0, 3, 1, 2)
        if self.training:
            self._nbt_pending += 1
        y = torch.nn.functional.batch_norm(
            x, self.running_mean, self.running_var, self.weight,
            self.bias, self.training, self.momentum, self.eps)
        if addend is not None:
            y = y + addend
        return torch.relu(y) if self.relu else y
