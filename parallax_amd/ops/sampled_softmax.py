"""Sampled-softmax loss (LM1B's 793k-way softmax with 8192 candidates).

Reference analog: TF's log-uniform candidate sampler + sampled_softmax_loss
over softmax_w [V, D] / softmax_b [V]
(examples/lm1b/language_model.py:41-50,95-100, num_sampled=8192).

The weight/bias rows are fetched through ``embedding_lookup`` so their
gradients arrive as indexed slices — on the full-vocab softmax table that
is exactly what makes the hybrid (sparse-PS) path pay: per step only
(batch*T + num_sampled) rows of the 793k-row table carry gradient.

The [B, D] x [S, D]^T candidate GEMM runs on hipBLASLt MFMA; the CE over
[B, 1+S] uses the fused softmax-CE kernel.
"""

import math

import torch

from parallax_amd.ops.embedding import embedding_lookup
from parallax_amd.ops.softmax_ce import softmax_cross_entropy


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

    true_w = rows(weight, labels)                     # [N, D]
    sampled_w = rows(weight, sampled_ids)             # [S, D]
    true_logits = (inputs * true_w).sum(-1)           # [N]
    sampled_logits = inputs @ sampled_w.t()           # [N, S]
    if bias is not None:
        true_logits = true_logits + embedding_lookup(bias, labels)
        sampled_logits = sampled_logits + embedding_lookup(bias, sampled_ids)

    # subtract log expected counts (TF sampled-softmax correction)
    true_logits = true_logits - torch.log(
        _expected_count(labels, num_classes, num_sampled).clamp_min(1e-20))
    sampled_logits = sampled_logits - torch.log(
        sampled_expected.clamp_min(1e-20))

    if remove_accidental_hits:
        hits = labels.unsqueeze(1) == sampled_ids.unsqueeze(0)  # [N, S]
        sampled_logits = sampled_logits.masked_fill(hits, -1e9)

    # The fused CE kernel accumulates max/sum-exp in fp32 internally, so
    # logits can stay in the GEMM dtype (bf16 under autocast) — avoids
    # materializing an [N, 1+S] fp32 copy (84 MB at the LM1B shape).
    logits = torch.cat([true_logits.unsqueeze(1).to(sampled_logits.dtype),
                        sampled_logits], dim=1)
    target = torch.zeros(labels.shape[0], dtype=torch.long, device=device)
    return softmax_cross_entropy(logits, target, reduction='none').float()
