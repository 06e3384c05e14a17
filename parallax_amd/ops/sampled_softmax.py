"""Sampled-softmax loss (LM1B's 793k-way softmax with 8192 candidates).

Reference analog: TF's log-uniform candidate sampler + sampled_softmax_loss
over softmax_w [V, D] / softmax_b [V]
(examples/lm1b/language_model.py:41-50,95-100, num_sampled=8192).

The weight/bias rows are fetched through ``embedding_lookup`` so their
gradients arrive as indexed slices — on the full-vocab softmax table that
is exactly what makes the hybrid (sparse-PS) path pay: per step only
(batch*T + num_sampled) rows of the 793k-row table carry gradient.

The [B, D] x [S, D]^T candidate GEMM runs on hipBLASLt MFMA; the CE +
per-row/column logit corrections run in ONE fused kernel (sampled_ce_*,
ops/hip/csrc/softmax_ce.hip) without materializing corrected copies of
the [B, S] matrix.
"""

import math

import torch

from parallax_amd import ops as _ops
from parallax_amd.ops.embedding import embedding_lookup
from parallax_amd.ops.softmax_ce import softmax_cross_entropy


def _ops_native():
    return _ops.native_available()


class _SampledCEFn(torch.autograd.Function):
    """Fused sampled-softmax CE (ops/hip/csrc/softmax_ce.hip):
    loss[i] = LSE_j([t_i+rc_i | s_ij+cc_j]) - (t_i + rc_i)."""

    @staticmethod
    def forward(ctx, true_logits, sampled_logits, row_corr, col_corr):
        loss, lse = _ops.ext().sampled_ce_fwd(true_logits, sampled_logits,
                                              row_corr, col_corr)
        ctx.save_for_backward(true_logits, sampled_logits, row_corr,
                              col_corr, lse)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        true_logits, sampled_logits, row_corr, col_corr, lse = \
            ctx.saved_tensors
        d_true, d_sampled = _ops.ext().sampled_ce_bwd(
            true_logits, sampled_logits, row_corr, col_corr, lse,
            dloss.contiguous())
        d_row = d_true.to(row_corr.dtype)
        d_col = d_sampled.sum(dim=0).to(col_corr.dtype)
        return d_true, d_sampled, d_row, d_col


def log_uniform_candidates(num_sampled, num_classes, device,
                           generator=None):
    """Log-uniform (Zipfian) candidate sampler (TF LogUniformCandidateSampler
    semantics): P(c) = log((c+2)/(c+1)) / log(V+1), sampled with
    replacement.  Returns (sampled_ids [S], sampled_expected [S])."""
    u = torch.rand(num_sampled, device=device, generator=generator)
    log_range = math.log(num_classes + 1)
    ids = (torch.exp(u * log_range) - 1).long().clamp_(0, num_classes - 1)
    probs = ((ids.float() + 2.0) / (ids.float() + 1.0)).log() / log_range
    expected = probs * num_sampled
    return ids, expected


def _expected_count(ids, num_classes, num_sampled):
    log_range = math.log(num_classes + 1)
    probs = ((ids.float() + 2.0) / (ids.float() + 1.0)).log() / log_range
    return probs * num_sampled


class _PxBtFn(torch.autograd.Function):
    """inputs @ w^T on the 8-phase MFMA GEMM (w is [S, D] row-major =
    the kernel's natural B operand); backward through library GEMMs
    (dw is a transposed-A split-K shape)."""

    @staticmethod
    def forward(ctx, a, b):
        ctx.save_for_backward(a, b)
        from parallax_amd import ops as _o
        return _o.ext().pxgemm8_bt(a.contiguous(), b.contiguous())

    @staticmethod
    def backward(ctx, g):
        a, b = ctx.saved_tensors
        return g @ b, g.transpose(0, 1) @ a


def _candidates_matmul(inputs, sampled_w):
    import os as _os
    from parallax_amd import ops as _o
    if (inputs.is_cuda and inputs.dtype == torch.bfloat16
            and sampled_w.dtype == torch.bfloat16
            and inputs.shape[1] % 64 == 0 and sampled_w.shape[0] % 16 == 0
            and _o.native_available()
            and hasattr(_o.ext(), 'pxgemm8_bt')
            and _os.environ.get('PARALLAX_PX_CANDIDATES', '1') == '1'):
        return _PxBtFn.apply(inputs, sampled_w)
    return inputs @ sampled_w.t()


def sampled_softmax_loss(weight, bias, labels, inputs, num_sampled,
                         num_classes, generator=None,
                         remove_accidental_hits=False):
    """weight: [V, D] Parameter OR a module with ``forward(ids) -> rows``
    (PartitionedEmbedding); bias: [V] Parameter or None; labels: [N] int64;
    inputs: [N, D].  Returns per-example losses [N]."""
    device = inputs.device
    sampled_ids, sampled_expected = log_uniform_candidates(
        num_sampled, num_classes, device, generator)

    def rows(w, ids):
        if isinstance(w, torch.nn.Module):
            return w(ids)
        return embedding_lookup(w, ids)

    # table rows are fp32; cast to the input dtype (no-op when equal) so
    # pure-bf16 bodies matmul without autocast — the .to() backward casts
    # the row grads back to fp32 before the indexed-slices grad forms
    true_w = rows(weight, labels).to(inputs.dtype)    # [N, D]
    sampled_w = rows(weight, sampled_ids).to(inputs.dtype)  # [S, D]
    true_logits = (inputs * true_w).sum(-1)           # [N]
    sampled_logits = _candidates_matmul(inputs, sampled_w)  # [N, S]

    # per-row / per-column logit corrections: + bias - log(expected_count)
    # (TF sampled-softmax correction); tiny [N] / [S] vectors
    row_corr = -torch.log(
        _expected_count(labels, num_classes, num_sampled).clamp_min(1e-20))
    col_corr = -torch.log(sampled_expected.clamp_min(1e-20))
    if bias is not None:
        row_corr = row_corr + embedding_lookup(bias, labels)
        col_corr = col_corr + embedding_lookup(bias, sampled_ids)

    if (not remove_accidental_hits and inputs.is_cuda
            and _ops_native()):
        # fused kernel: CE over [true+row_corr | sampled+col_corr] without
        # materializing corrected copies of the [N, S] matrix.  Note:
        # autocast computes the row-wise sum in fp32, so true_logits must
        # be cast to the GEMM dtype the kernel dispatches on.
        return _SampledCEFn.apply(true_logits.to(sampled_logits.dtype),
                                  sampled_logits.contiguous(),
                                  row_corr, col_corr)

    true_logits = true_logits + row_corr.to(true_logits.dtype)
    sampled_logits = sampled_logits + col_corr.to(sampled_logits.dtype)
    if remove_accidental_hits:
        hits = labels.unsqueeze(1) == sampled_ids.unsqueeze(0)  # [N, S]
        sampled_logits = sampled_logits.masked_fill(hits, -1e9)

    logits = torch.cat([true_logits.unsqueeze(1).to(sampled_logits.dtype),
                        sampled_logits], dim=1)
    target = torch.zeros(labels.shape[0], dtype=torch.long, device=device)
    return softmax_cross_entropy(logits, target, reduction='none').float()
