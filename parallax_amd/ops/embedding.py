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


class _ShardedLookup(torch.autograd.Function):
    """Cross-rank row fetch for shard-only storage
    (PSConfig.replicate_variables=False, reference
    ps/between_graph_parallel.py: variables live ONLY on their PS task).

    forward: ids are grouped by owner rank, exchanged all-to-all, each
    owner gathers its local rows and sends them back.  backward: the grad
    rows are pushed to their owners the same way; the returned grad is a
    sparse COO over the LOCAL shard (local row indices) which the sparse
    engine applies WITHOUT any further exchange (it is already the global
    contribution set for this rank's rows).

    Every rank must run the same lookups in the same order (data-parallel
    lockstep) — the collectives are symmetric by construction.
    """

    @staticmethod
    def forward(ctx, local_weight, flat_ids, module):
        comm = module._comm
        world = comm.world_size
        with torch.no_grad():
            owner = module.owner_of(flat_ids)
            order = torch.argsort(owner, stable=True)
            sorted_ids = flat_ids[order]
            counts = torch.bincount(owner, minlength=world)
            counts_l = [int(c) for c in counts.tolist()]
            send = list(torch.split(sorted_ids, counts_l))
            recv_ids = comm.all_to_all_v(
                [s.unsqueeze(1) for s in send])
            recv_ids = [r.squeeze(1) for r in recv_ids]
            rows_back = [local_weight[module.local_index(r)]
                         for r in recv_ids]
            got = comm.all_to_all_v(rows_back)
            out = torch.empty(flat_ids.shape[0], module.embedding_dim,
                              dtype=local_weight.dtype,
                              device=local_weight.device)
            out[order] = torch.cat(got)
        ctx.save_for_backward(order, *recv_ids)
        ctx.module = module
        ctx.counts_l = counts_l
        return out

    @staticmethod
    def backward(ctx, grad_out):
        order, *recv_ids = ctx.saved_tensors
        module = ctx.module
        comm = module._comm
        sorted_grads = grad_out.contiguous()[order]
        send = list(torch.split(sorted_grads, ctx.counts_l))
        got_rows = comm.all_to_all_v(send)
        local_ids = module.local_index(torch.cat(recv_ids)) \
            if recv_ids else torch.zeros(0, dtype=torch.long,
                                         device=grad_out.device)
        rows = torch.cat(got_rows) if got_rows else \
            grad_out.new_zeros((0, module.embedding_dim))
        grad_w = torch.sparse_coo_tensor(
            local_ids.reshape(1, -1),
            rows.to(module.weight.dtype),
            (module._local_rows, module.embedding_dim),
            check_invariants=False)
        return grad_w, None, None


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

    ``to_sharded(comm)`` (PSConfig.replicate_variables=False) switches to
    SHARD-ONLY storage for tables that exceed replication budget: rank r
    keeps only partitions k with k % world == r; lookups fetch rows from
    their owners over an all-to-all and gradients are pushed back the same
    way (_ShardedLookup).  hipGraph capture is unavailable in this mode
    (collectives inside forward).
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
        self.weight._parallax_module = self
        self._offsets_t = None
        self._sharded = False
        self._comm = None
        self._owned = None          # partition indices this rank stores
        self._local_rows = 0
        self._local_base_t = None   # per-partition local_start - offset

    def _offsets_tensor(self, device):
        if self._offsets_t is None or self._offsets_t.device != device:
            self._offsets_t = torch.tensor(self.offsets, dtype=torch.long,
                                           device=device)
        return self._offsets_t

    def shard_of(self, flat_ids):
        """shard index for each id (searchsorted over row offsets)."""
        offs = self._offsets_tensor(flat_ids.device)
        return torch.searchsorted(offs, flat_ids, right=True) - 1

    # -- shard-only storage (PSConfig.replicate_variables=False) ------------
    def to_sharded(self, comm, part_to_rank=None):
        """Drop every row this rank does not own.  ``part_to_rank`` maps
        partition k -> owning rank; the default is round-robin k % world.
        The sparse engine passes a byte-size greedy least-loaded placement
        (reference GreedyLoadBalancingStrategy,
        ps/between_graph_parallel.py:30-70).  Must run AFTER the rank-0
        broadcast (all ranks hold the same full table) and BEFORE the
        first step; the placement must be computed identically on every
        rank (it is deterministic in registration order)."""
        if self._sharded:
            return
        world, rank = comm.world_size, comm.rank
        dev = self.weight.device
        if part_to_rank is None:
            part_to_rank = [k % world for k in range(self.num_partitions)]
        assert len(part_to_rank) == self.num_partitions
        owned = [k for k in range(self.num_partitions)
                 if part_to_rank[k] == rank]
        pieces, local_base, local_off = [], [-1] * self.num_partitions, 0
        for k in owned:
            off, size = self.offsets[k], self.sizes[k]
            pieces.append(self.weight.data[off:off + size].clone())
            local_base[k] = local_off - off
            local_off += size
        local = torch.cat(pieces) if pieces else \
            self.weight.data.new_zeros((0, self.embedding_dim))
        self.weight.data = local.contiguous()
        self._sharded = True
        self._comm = comm
        self._owned = owned
        self._part_to_rank = list(part_to_rank)
        self._part_rank_t = torch.tensor(part_to_rank, dtype=torch.long,
                                         device=dev)
        self._local_rows = local.shape[0]
        self._local_base_t = torch.tensor(local_base, dtype=torch.long,
                                          device=dev)

    def owner_of(self, flat_ids):
        """Owning rank for each id (partition -> rank placement map)."""
        return self._part_rank_t[self.shard_of(flat_ids)]

    def local_index(self, flat_ids):
        """Global id -> row index in THIS rank's local shard storage
        (valid only for ids this rank owns)."""
        part = self.shard_of(flat_ids)
        return flat_ids + self._local_base_t[part]

    def forward(self, ids):
        if self._sharded:
            flat = ids.reshape(-1)
            out = _ShardedLookup.apply(self.weight, flat, self)
            return out.view(*ids.shape, self.embedding_dim)
        return embedding_lookup(self.weight, ids)

    def full_weight(self):
        """The logical [V, D] table.  In sharded mode this is a COLLECTIVE
        (allgather of shards) and must be called on every rank."""
        if not self._sharded:
            return self.weight.detach()
        world = self._comm.world_size
        gathered = self._comm.all_gather_v(self.weight.detach())
        full = self.weight.new_empty((self.num_embeddings,
                                      self.embedding_dim))
        for r in range(world):
            local_off = 0
            for k in range(self.num_partitions):
                if self._part_to_rank[k] != r:
                    continue
                off, size = self.offsets[k], self.sizes[k]
                full[off:off + size] = gathered[r][local_off:local_off
                                                   + size]
                local_off += size
        return full

    # --- checkpoint layout parity (reference 'emb/part_k' shards) ----------
    def _save_to_state_dict(self, destination, prefix, keep_vars):
        if self._sharded:
            # collective: every rank contributes its shards; the emitted
            # parts.k keep the replicated layout so a sharded run's
            # checkpoint restores under ANY world size / mode
            w = self.full_weight()
        else:
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
                    self._copy_full(state_dict[prefix + 'weight'])
                return
            missing_keys.append(prefix + 'parts.0')
            return
        with torch.no_grad():
            full = torch.cat([state_dict[k].to(self.weight.dtype)
                              for k in part_keys], dim=0)
            if full.shape[0] != self.num_embeddings:
                error_msgs.append(
                    'PartitionedEmbedding %s: checkpoint rows %s != %s'
                    % (prefix, tuple(full.shape),
                       (self.num_embeddings, self.embedding_dim)))
                return
            self._copy_full(full)

    def _copy_full(self, full):
        if not self._sharded:
            self.weight.copy_(full)
            return
        local_off = 0
        for k in self._owned:
            off, size = self.offsets[k], self.sizes[k]
            self.weight.data[local_off:local_off + size].copy_(
                full[off:off + size])
            local_off += size
