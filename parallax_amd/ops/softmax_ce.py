"""Fused softmax cross-entropy with sparse (integer) labels.

Reference analog: TF softmax_cross_entropy_with_logits CUDA kernels used by
the ResNet loss and NMT (benchmark_cnn.py:290-302).

GPU: hand-written HIP kernel (ops/hip/softmax_ce.hip) — one pass computing
row max, sum-exp and the picked logit; backward is the fused
(softmax - onehot) * dloss kernel.  CPU: torch reference.
"""

import torch

from parallax_amd import ops as _ops


class _SoftmaxCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        loss, lse = _ops.ext().softmax_ce_fwd(logits, labels)
        ctx.save_for_backward(logits, labels, lse)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, labels, lse = ctx.saved_tensors
        dlogits = _ops.ext().softmax_ce_bwd(logits, labels, lse,
                                            dloss.contiguous())
        return dlogits, None


def softmax_cross_entropy(logits, labels, reduction='mean'):
    """loss[i] = -log softmax(logits[i])[labels[i]]; logits [N, C],
    labels [N] int64."""
    if _ops.use_native(logits):
        losses = _SoftmaxCEFn.apply(logits.contiguous(), labels.contiguous())
    else:
        losses = torch.nn.functional.cross_entropy(
            logits.float(), labels, reduction='none').to(logits.dtype)
    if reduction == 'mean':
        return losses.mean()
    if reduction == 'sum':
        return losses.sum()
    return losses
