"""Sparse (indexed-slices) gradient engines.

Two modes, mirroring the reference's architecture split:

* SparseAllGatherEngine — MPI mode: each rank keeps a full replica of every
  sparse table; per step each rank DEDUPS its own indexed-slices grad
  (segment-sum kernel, contract N2), all ranks ALLGATHER (ids, rows)
  (contract N4; reference mpi/graph_transform.py:46-61 IndexedSlices ->
  values/indices allgather), re-dedup the union, and every rank applies the
  identical sparse row update.  ``average_sparse`` divides by occurrence
  count (reference config.py:134-137, take_indexed_slices_grad
  average_option, graph_transform_lib.py:385-390).

* SparsePSEngine — PS/HYBRID mode: the in-process parameter server
  (contract N7).  Each table shard has an OWNER rank; shards live in the
  owner's HBM3E (288 GB/GPU -> whole 800k-vocab tables fit trivially, so
  non-owners hold a read replica refreshed after update — the reference's
  ``replicate_variables`` mirror, graph_transform_lib.py:584-704).
  Per step: local dedup (the reference's machine-local aggregation,
  graph_transform_lib.py:1372-1556, is implicit on one node) -> RCCL
  all-to-all routes (ids, grad rows) to owners over xGMI -> owner
  segment-sums the union and applies SparseApplyAdagrad-etc on its shard
  -> owner broadcasts the touched rows back to the mirrors (the mirror
  refresh, graph_transform_lib.py:406-415).

Gradient capture: the engines intercept sparse COO grads from autograd via
per-parameter hooks; param.grad is left None for sparse params (the
optimizer's dense path never sees them).
"""

import torch
import torch.distributed as dist

from parallax_amd.lib import parallax_log
from parallax_amd.ops.sparse import unique_sum_rows


class _SparseGradSlot(object):
    __slots__ = ('info', 'ids', 'rows')

    def __init__(self, info):
        self.info = info
        self.ids = None
        self.rows = None


def _capture_hook(slot):
    def hook(grad):
        # grad is a sparse COO (indices [1, n], values [n, d]); stash it for
        # the engine and replace it with an EMPTY sparse grad so autograd's
        # param.grad accumulation and the optimizer's own sparse path both
        # see nothing (the engine applies the update itself).
        g = grad
        if g.layout == torch.sparse_coo:
            ids = g._indices().flatten()
            rows = g._values()
        else:
            # a dense grad reached a sparse-tagged param (e.g. weight reuse
            # in a dense matmul): treat all rows as touched
            ids = torch.arange(g.shape[0], device=g.device)
            rows = g
        if rows.dim() == 1:  # 1-D table (bias): keep rows 2-D internally
            rows = rows.unsqueeze(1)
        if slot.ids is None:
            slot.ids, slot.rows = ids, rows
        else:  # several uses of the table in one step (e.g. sampled softmax)
            slot.ids = torch.cat([slot.ids, ids])
            slot.rows = torch.cat([slot.rows, rows])
        return torch.sparse_coo_tensor(
            torch.zeros((1, 0), dtype=torch.long, device=grad.device),
            torch.zeros((0,) + tuple(grad.shape[1:]), dtype=grad.dtype,
                        device=grad.device),
            grad.shape, check_invariants=False)

    return hook


class _SparseEngineBase(object):
    def __init__(self, comm, sparse_infos, optimizer, average_sparse=False):
        self.comm = comm
        self.infos = list(sparse_infos)
        self.optimizer = optimizer
        self.average_sparse = average_sparse
        self.slots = [_SparseGradSlot(gi) for gi in self.infos]
        self._hooks = []
        for slot in self.slots:
            h = slot.info.param.register_hook(_capture_hook(slot))
            self._hooks.append(h)

    def prepare_step(self):
        for s in self.slots:
            s.ids = None
            s.rows = None

    def close(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []

    def _apply_rows(self, param, ids, rows):
        if param.dim() == 1 and rows.dim() == 2:
            rows = rows.squeeze(1)
        self.optimizer.sparse_row_step(param, ids, rows)


class SparseAllGatherEngine(_SparseEngineBase):
    """MPI-mode sparse path (replicated tables + allgather)."""

    def reduce_and_apply(self):
        for s in self.slots:
            if s.ids is None:
                continue
            # 1. local dedup (segment-sum kernel)
            u_ids, u_rows, _ = unique_sum_rows(s.ids, s.rows)
            # 2. allgather (ids, rows) across ranks (contract N4)
            if self.comm.world_size > 1:
                all_ids = torch.cat(self.comm.all_gather_v(u_ids))
                all_rows = torch.cat(self.comm.all_gather_v(u_rows))
            else:
                all_ids, all_rows = u_ids, u_rows
            # 3. global dedup (+ optional count-average) and apply
            g_ids, g_rows, counts = unique_sum_rows(
                all_ids, all_rows, average=self.average_sparse)
            self._apply_rows(s.info.param, g_ids, g_rows)


class SparsePSEngine(_SparseEngineBase):
    """HYBRID/PS-mode sparse path (in-process PS, sharded ownership).

    Ownership: parameter k (in registration order) is owned by rank
    ``k % world_size`` — for a PartitionedEmbedding's p shards this spreads
    shards round-robin over ranks, the greedy-load-balance analog
    (reference ps/between_graph_parallel.py:30-70) for equal-size fixed
    partitions.
    """

    def __init__(self, comm, sparse_infos, optimizer, average_sparse=False,
                 replicate_variables=True):
        super().__init__(comm, sparse_infos, optimizer, average_sparse)
        self.replicate = replicate_variables
        self.owner = {}
        for k, gi in enumerate(self.infos):
            self.owner[id(gi.param)] = k % comm.world_size

    def reduce_and_apply(self):
        ws = self.comm.world_size
        for s in self.slots:
            if s.ids is None:
                continue
            p = s.info.param
            owner = self.owner[id(p)]
            # 1. local dedup (machine-local aggregation collapses to this on
            #    one node — reference graph_transform_lib.py:1404-1476)
            u_ids, u_rows, u_counts = unique_sum_rows(s.ids, s.rows)
            if ws == 1:
                if self.average_sparse:
                    u_rows = u_rows / u_counts.unsqueeze(1).to(u_rows.dtype)
                self._apply_rows(p, u_ids, u_rows)
                continue
            # 2. push to owner: gather_v is cheapest for single-owner tables
            #    (every rank sends to one destination)
            all_ids = self.comm.all_gather_v(u_ids)
            all_rows = self.comm.all_gather_v(u_rows)
            all_counts = self.comm.all_gather_v(u_counts)
            if self.comm.rank == owner:
                g_ids = torch.cat(all_ids)
                g_rows = torch.cat(all_rows)
                g_ids, g_rows, _ = unique_sum_rows(g_ids, g_rows)
                if self.average_sparse:
                    cat_ids = torch.cat(all_ids)
                    cat_counts = torch.cat(all_counts).to(torch.float32)
                    # total occurrence count per unique id
                    _, tot, _ = unique_sum_rows(cat_ids,
                                                cat_counts.unsqueeze(1))
                    g_rows = g_rows / tot.to(g_rows.dtype)
                self._apply_rows(p, g_ids, g_rows)
            # 3. mirror refresh: owner broadcasts updated rows for the
            #    union of touched ids (reference mirror update,
            #    graph_transform_lib.py:406-415,547-557)
            if self.replicate:
                cat_ids = torch.cat(all_ids)
                touched = torch.unique(cat_ids)
                rows = p.data.index_select(0, touched)
                dist.broadcast(rows, src=owner)
                if self.comm.rank != owner:
                    p.data.index_copy_(0, touched, rows)

    def owned_params(self):
        return [gi.param for gi in self.infos
                if self.owner[id(gi.param)] == self.comm.rank]
