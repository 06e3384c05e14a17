"""Embedding lookup producing indexed-slices gradients, plus the
fixed-size-partitioned variant.

Reference behavior being reproduced:
  * lookup of [B, T] ids in a [V, D] table -> [B, T, D]
    (examples/lm1b/language_model.py:26-40, nmt/model_helper.py:310-311);
  * the gradient w.r.t. the table is an IndexedSlices (values = upstream
    grad rows, indices = the ids) — the thing the tagging pass (N1)
    classifies as SPARSE;
  * ``tf.fixed_size_partitioner`` splits big tables row-wise into p shards
    named ``<name>/part_k`` (partitions.py:35-51); checkpoint layout keeps
    those shard names (§5.4 of SURVEY).

MI355X-native design: one nn.Module whose weight gradient is a torch sparse
COO tensor (duplicate indices allowed; dedup happens in the engine's
segment-sum kernel, contract N2).  The partitioned variant holds p shard
Parameters so the sparse engines can place/exchange shards independently.
"""

import torch
import torch.nn as nn

from parallax_amd import ops as _ops
from parallax_amd.ops.sparse import rows_gather


class _EmbeddingLookup(torch.autograd.Function):
    """Gather rows; backward emits a sparse COO grad (indexed slices)."""

    @staticmethod
    def forward(ctx, weight, flat_ids):
        ctx.save_for_backward(flat_ids)
        ctx.vocab = weight.shape[0]
        ctx.dim = weight.shape[1]
        ctx.w_dtype = weight.dtype
        return rows_gather(weight, flat_ids)

    @staticmethod
    def backward(ctx, grad_out):
        (flat_ids,) = ctx.saved_tensors
        # IndexedSlices: values = grad rows, indices = ids (duplicates kept;
        # engines dedup with the segment-sum kernel).
        grad_values = grad_out.reshape(-1, ctx.dim)
        grad_w = torch.sparse_coo_tensor(
            flat_ids.reshape(1, -1), grad_values.to(ctx.w_dtype),
            (ctx.vocab, ctx.dim), check_invariants=False)
        return grad_w, None


class _VectorLookup(torch.autograd.Function):
    """1-D table lookup (bias vectors) with a sparse [V] grad."""

    @staticmethod
    def forward(ctx, weight, flat_ids):
        ctx.save_for_backward(flat_ids)
        ctx.vocab = weight.shape[0]
        ctx.w_dtype = weight.dtype
        return weight.index_select(0, flat_ids)

    @staticmethod
    def backward(ctx, grad_out):
        (flat_ids,) = ctx.saved_tensors
        grad_w = torch.sparse_coo_tensor(
            flat_ids.reshape(1, -1), grad_out.reshape(-1).to(ctx.w_dtype),
            (ctx.vocab,), check_invariants=False)
        return grad_w, None


def embedding_lookup(weight, ids):
    """Functional lookup with indexed-slices grad w.r.t. weight.
    weight may be [V, D] (row lookup) or [V] (bias lookup)."""
    flat = ids.reshape(-1)
    if weight.dim() == 1:
        return _VectorLookup.apply(weight, flat).view(*ids.shape)
    out = _EmbeddingLookup.apply(weight, flat)
    return out.view(*ids.shape, weight.shape[1])


class ParallaxEmbedding(nn.Module):
    """Single-table embedding with indexed-slices gradients."""

    _parallax_is_embedding = True

    def __init__(self, num_embeddings, embedding_dim, dtype=torch.float32,
                 init_scale=None):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        w = torch.empty(num_embeddings, embedding_dim, dtype=dtype)
        scale = init_scale if init_scale is not None else embedding_dim ** -0.5
        nn.init.uniform_(w, -scale, scale)
        self.weight = nn.Parameter(w)

    def forward(self, ids):
        return embedding_lookup(self.weight, ids)


class PartitionedEmbedding(nn.Module):
    """Row-partitioned embedding: p shard Parameters ``part_0..part_{p-1}``
    (reference fixed_size_partitioner shard naming, partitions.py:51;
    checkpoint layout parity §5.4).

    Lookup routes each id to its shard (first ``V % p`` shards hold one
    extra row) and re-assembles the output.  Each shard's gradient is its
    own indexed-slices, so the sparse engine exchanges/updates shards
    independently — that is what partition search varies.
    """

    _parallax_is_embedding = True

    def __init__(self, num_embeddings, embedding_dim, partitioner,
                 dtype=torch.float32, init_scale=None):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        bounds = partitioner(num_embeddings)  # [(offset, size)]
        self.num_partitions = len(bounds)
        self.offsets = [o for o, _ in bounds]
        self.sizes = [s for _, s in bounds]
        scale = init_scale if init_scale is not None else embedding_dim ** -0.5
        self.parts = nn.ParameterList()
        for k, (off, size) in enumerate(bounds):
            w = torch.empty(size, embedding_dim, dtype=dtype)
            nn.init.uniform_(w, -scale, scale)
            self.parts.append(nn.Parameter(w))
        for k, p in enumerate(self.parts):
            p._parallax_partition = (k, self.num_partitions,
                                     self.offsets[k], num_embeddings)
        self._offsets_t = None

    def _offsets_tensor(self, device):
        if self._offsets_t is None or self._offsets_t.device != device:
            self._offsets_t = torch.tensor(self.offsets, dtype=torch.long,
                                           device=device)
        return self._offsets_t

    def shard_of(self, flat_ids):
        """shard index for each id (searchsorted over row offsets)."""
        offs = self._offsets_tensor(flat_ids.device)
        return torch.searchsorted(offs, flat_ids, right=True) - 1

    def forward(self, ids):
        flat = ids.reshape(-1)
        shard_ids = self.shard_of(flat)
        out = None
        for k in range(self.num_partitions):
            mask = shard_ids == k
            if not torch.any(mask):
                continue
            pos = torch.nonzero(mask).flatten()
            local = flat[pos] - self.offsets[k]
            rows = embedding_lookup(self.parts[k], local)
            if out is None:
                out = rows.new_zeros(flat.shape[0], self.embedding_dim)
            out = out.index_copy(0, pos, rows)
        if out is None:
            out = self.parts[0].new_zeros(flat.shape[0], self.embedding_dim)
        return out.view(*ids.shape, self.embedding_dim)

    def full_weight(self):
        """Materialize the logical [V, D] table (tests / checkpoint export)."""
        return torch.cat([p.detach() for p in self.parts], dim=0)
