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
segment-sum kernel, contract N2).  The partitioned variant keeps ONE
resident table with VIRTUAL shard metadata — see PartitionedEmbedding.
"""

import torch
import torch.nn as nn

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
    """Row-partitioned embedding with VIRTUAL shards.

    Reference behavior: ``tf.fixed_size_partitioner`` splits the table into
    p shard variables ``<name>/part_k`` placed on different PS tasks
    (partitions.py:35-51); checkpoints keep the shard layout (§5.4).

    MI355X-native design: 288 GB of HBM3E per GPU makes physical sharding
    of storage pointless on one node — the table is ONE resident Parameter
    (lookup = one gather kernel, gradient = one indexed-slices tensor) and
    the partitioning is METADATA (`weight._parallax_partition_meta` =
    (num_partitions, offsets)) consumed by
      * the sparse engines (shard granularity of the exchange/update and of
        partition search),
      * the checkpoint layer (state_dict emits/accepts ``parts.k`` shard
        slices, so the on-disk layout keeps reference shard structure and a
        checkpoint written with p shards restores under any p').
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
        w = torch.empty(num_embeddings, embedding_dim, dtype=dtype)
        nn.init.uniform_(w, -scale, scale)
        self.weight = nn.Parameter(w)
        self.weight._parallax_partition_meta = (self.num_partitions,
                                                tuple(self.offsets))
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
        return embedding_lookup(self.weight, ids)

    def full_weight(self):
        """The logical [V, D] table."""
        return self.weight.detach()

    # --- checkpoint layout parity (reference 'emb/part_k' shards) ----------
    def _save_to_state_dict(self, destination, prefix, keep_vars):
        w = self.weight if keep_vars else self.weight.detach()
        for k in range(self.num_partitions):
            off, size = self.offsets[k], self.sizes[k]
            destination[prefix + 'parts.%d' % k] = w[off:off + size]

    def _load_from_state_dict(self, state_dict, prefix, local_metadata,
                              strict, missing_keys, unexpected_keys,
                              error_msgs):
        part_keys = sorted(
            (k for k in state_dict if k.startswith(prefix + 'parts.')),
            key=lambda k: int(k.rsplit('.', 1)[1]))
        if not part_keys:
            if prefix + 'weight' in state_dict:  # plain layout also accepted
                with torch.no_grad():
                    self.weight.copy_(state_dict[prefix + 'weight'])
                return
            missing_keys.append(prefix + 'parts.0')
            return
        with torch.no_grad():
            full = torch.cat([state_dict[k].to(self.weight.dtype)
                              for k in part_keys], dim=0)
            if full.shape != self.weight.shape:
                error_msgs.append(
                    'PartitionedEmbedding %s: checkpoint rows %s != %s'
                    % (prefix, tuple(full.shape), tuple(self.weight.shape)))
                return
            self.weight.copy_(full)
