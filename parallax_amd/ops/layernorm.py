"""LayerNorm — hand-written CDNA4 HIP kernel on GPU, torch reference on CPU.

Reference analog: TF fused layer_norm used optionally by the NMT LSTM
variants.  GPU kernel: ops/hip/layernorm.hip (one workgroup
per row-block, wave64 shuffle reductions, bf16x8 vector loads per
guide G13).
"""

import torch
import torch.nn as nn

from parallax_amd import ops as _ops


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        y, mean, rstd = _ops.ext().layernorm_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = _ops.ext().layernorm_bwd(dy.contiguous(), x, weight,
                                              mean, rstd)
        return dx, dw, db, None


def layer_norm(x, normalized_shape, weight, bias, eps=1e-5):
    if _ops.use_native(x):
        shp = x.shape
        x2 = x.reshape(-1, shp[-1]).contiguous()
        # the kernel reads weight/bias with x's element type: cast when
        # the params are fp32 over a bf16 body (autocast mode) — the
        # .to() backward returns the [d] grads to fp32 (differentiable)
        w = weight.to(x.dtype) if weight.dtype != x.dtype else weight
        b = bias.to(x.dtype) if bias.dtype != x.dtype else bias
        y = _LayerNormFn.apply(x2, w, b, eps)
        return y.view(shp)
    return torch.nn.functional.layer_norm(x, normalized_shape, weight, bias,
                                          eps)


class ParallaxLayerNorm(nn.Module):
    def __init__(self, dim, eps=1e-5, dtype=torch.float32):
        super().__init__()
        self.dim = (dim,) if isinstance(dim, int) else tuple(dim)
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(self.dim, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(self.dim, dtype=dtype))

    def forward(self, x):
        return layer_norm(x, self.dim, self.weight, self.bias, self.eps)
