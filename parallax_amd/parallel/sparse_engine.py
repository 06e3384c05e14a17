"""Sparse (indexed-slices) gradient engines.

Two modes, mirroring the reference's architecture split:

* SparseAllGatherEngine — MPI mode: Horovod-style IndexedSlices handling
  (contract N4; reference mpi/graph_transform.py:46-61): each rank
  ALLGATHERS its RAW (values, indices) — no local aggregation, exactly
  Horovod's semantics — then every rank dedups the union (segment-sum
  kernel, contract N2) and applies the identical sparse row update.
  ``average_sparse`` divides by occurrence count (reference
  config.py:134-137, take_indexed_slices_grad average_option,
  graph_transform_lib.py:385-390).

* SparsePSEngine — PS/HYBRID mode: the in-process parameter server
  (contract N7), mapped to one MI355X node: each rank LOCALLY AGGREGATES
  its indexed slices first (the reference's machine-local aggregation,
  graph_transform_lib.py:1372-1556, PSConfig.local_aggregation), then the
  deduplicated (unique ids, summed rows) are allgathered over xGMI —
  message size ∝ unique ids, the whole point of the hybrid path — the
  union is segment-summed once more and the row update applied to the
  (replicated, 288 GB HBM3E makes replication free on one node) table on
  every rank, so tables and optimizer state stay bit-identical with no
  mirror-refresh broadcast.  The table's PARTITION metadata (virtual
  shards) sets the granularity partition search varies and the
  checkpoint-shard layout.  ``replicate_variables=False`` (the
  reference's shard-only PS placement) is ALSO supported for tables that
  exceed the replication budget: each rank stores only partitions
  k % world == rank, lookups fetch rows from their owners over an
  all-to-all (embedding._ShardedLookup) and the grad push rides the
  lookup's backward, so reduce_and_apply for those tables is purely
  local.  On one xGMI node the replicated default is faster (no
  per-lookup exchange); sharding is for capacity.

Gradient capture: the engines intercept sparse COO grads from autograd via
per-parameter hooks; the hook swallows the grad (replaces it with an empty
sparse tensor) so neither autograd accumulation nor the optimizer's own
sparse path sees it — the engine applies the update itself.
"""

import os

import torch
import torch.distributed as dist

from parallax_amd.lib import parallax_log
from parallax_amd.ops.sparse import (unique_sum_rows,
                                     unique_sum_rows_padded,
                                     _sorted_segments_presort)


class _SparseGradSlot(object):
    __slots__ = ('info', 'ids', 'rows')

    def __init__(self, info):
        self.info = info
        self.ids = None
        self.rows = None


def _capture_hook(slot):
    def hook(grad):
        g = grad
        dense_use = g.layout != torch.sparse_coo
        if dense_use:
            # a dense grad reached a sparse-tagged param (e.g. weight
            # reuse in a dense matmul): treat all rows as touched
            ids = torch.arange(g.shape[0], device=g.device)
            rows = g
        else:
            ids = g._indices().flatten()
            rows = g._values()
        if rows.dim() == 1:  # 1-D table (bias): keep rows 2-D internally
            rows = rows.unsqueeze(1)
        if slot.ids is None:
            slot.ids, slot.rows = ids, rows
        else:  # several uses of the table in one step (e.g. sampled softmax)
            slot.ids = torch.cat([slot.ids, ids])
            slot.rows = torch.cat([slot.rows, rows])
        if dense_use:
            # autograd forbids changing the grad LAYOUT in a hook, so the
            # swallowed replacement must stay dense: a zero tensor tagged
            # for the optimizer to skip (the engine applies the captured
            # rows itself; ParallaxOptimizer.step honors the tag)
            z = torch.zeros_like(g)
            z._parallax_swallowed = True
            return z
        return torch.sparse_coo_tensor(
            torch.zeros((1, 0), dtype=torch.long, device=grad.device),
            torch.zeros((0,) + tuple(grad.shape[1:]), dtype=grad.dtype,
                        device=grad.device),
            grad.shape, check_invariants=False)

    return hook


class _SparseEngineBase(object):
    def __init__(self, comm, sparse_infos, optimizer, average_sparse=False,
                 comm_dtype=torch.bfloat16):
        self.comm = comm
        self.infos = list(sparse_infos)
        self.optimizer = optimizer
        self.average_sparse = average_sparse
        # bf16 payloads only on the RCCL/GPU path; CPU/gloo plumbing tests
        # keep native dtype (and exact numerics)
        if comm.device.type != 'cuda':
            comm_dtype = None
        self.comm_dtype = comm_dtype
        self.slots = [_SparseGradSlot(gi) for gi in self.infos]
        self._hooks = []
        for slot in self.slots:
            h = slot.info.param.register_hook(_capture_hook(slot))
            self._hooks.append(h)
        # sync-free padded path (fixed shapes, device-side valid count):
        # removes every per-table host stall and makes the sparse tail
        # hipGraph-capturable.  GPU + padded-capable optimizer only; at
        # world>1 additionally requires every rank to contribute the SAME
        # id count each step (verified ONCE, below).
        env = os.environ.get('PARALLAX_PADDED_SPARSE', '1')
        self._padded = ((comm.device.type == 'cuda' or env == 'force')
                        and env != '0'
                        and getattr(optimizer, 'supports_padded_sparse',
                                    lambda: False)())
        self._counts_static = None  # None = unverified; True/False after

    def prepare_step(self):
        for s in self.slots:
            s.ids = None
            s.rows = None

    def capture_safe(self):
        """True when reduce_and_apply is hipGraph-capturable: the padded
        sync-free path with no collectives (world 1)."""
        return self._padded and self.comm.world_size == 1

    def _verify_static_counts(self, live):
        """One-time check (one sync) that every rank sees identical
        per-slot id counts — the requirement for the fixed-shape padded
        exchange.  Models with data-dependent lookup counts (bucketed
        batching) fall back to the dynamic-size path permanently."""
        if self._counts_static is not None:
            return self._counts_static
        counts = torch.tensor([s.ids.numel() for s in live]
                              + [len(live)],
                              dtype=torch.long, device=self.comm.device)
        lo = counts.clone()
        hi = counts.clone()
        dist.all_reduce(lo, op=dist.ReduceOp.MIN)
        dist.all_reduce(hi, op=dist.ReduceOp.MAX)
        self._counts_static = bool(torch.equal(lo, hi))
        if not self._counts_static:
            parallax_log.info(
                'sparse id counts differ across ranks: using the '
                'dynamic-size exchange (padded path disabled)')
        return self._counts_static

    def close(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []

    def _apply_rows(self, param, ids, rows, nvalid=None):
        if param.dim() == 1 and rows.dim() == 2:
            rows = rows.squeeze(1)
        self.optimizer.sparse_row_step(param, ids, rows, nvalid=nvalid)

    def _padded_exchange_and_apply(self, slot, u_ids, u_rows, u_counts,
                                   count_avg):
        """Fixed-shape exchange: every rank contributes exactly
        len(u_ids) entries (tail = id 0 / zero rows), so the allgather
        needs no counts exchange, no padding to max and no host syncs.
        ``u_counts`` carries per-id multiplicities (None = raw,
        multiplicity 1 each)."""
        send_rows = u_rows.to(self.comm_dtype) \
            if self.comm_dtype is not None else u_rows
        all_ids = self.comm.all_gather_fixed(u_ids)
        all_rows = self.comm.all_gather_fixed(
            send_rows.contiguous()).to(u_rows.dtype)
        in_counts = None
        if count_avg:
            in_counts = self.comm.all_gather_fixed(
                u_counts if u_counts is not None
                else torch.ones_like(u_ids, dtype=torch.float32))
        g_ids, g_rows, g_counts, nvalid = unique_sum_rows_padded(
            all_ids, all_rows, average=count_avg, in_counts=in_counts)
        if self.average_sparse and not count_avg:  # Horovod world-average
            g_rows = g_rows / float(self.comm.world_size)
        self._apply_rows(slot.info.param, g_ids, g_rows, nvalid=nvalid)

    # How ``average_sparse`` divides: the PS engine count-averages by
    # per-id occurrence (reference take_indexed_slices_grad
    # SPARSE_AVERAGE_BY_COUNTER, graph_transform_lib.py:385-390); the MPI
    # engine divides by WORLD SIZE (Horovod hvd.allreduce(IndexedSlices,
    # average=True), mpi/graph_transform.py:51-53) — the two reference
    # paths genuinely differ, so the engines do too.
    _average_mode = 'count'

    def _exchange_and_apply(self, slot, u_ids, u_rows, u_counts):
        """Allgather (ids, rows[, counts]) across ranks, dedup the union
        (optionally averaged, see _average_mode) and apply on every rank.
        Row payloads travel in ``comm_dtype`` (default bf16 — halves the
        xGMI bytes; the reduction itself stays fp32 in the segment-sum
        kernel)."""
        count_avg = self.average_sparse and self._average_mode == 'count'
        if self.comm.world_size > 1:
            send_rows = u_rows.to(self.comm_dtype) \
                if self.comm_dtype is not None else u_rows
            tensors = [u_ids, send_rows]
            if count_avg:
                tensors.append(u_counts)
            gathered = self.comm.all_gather_v_multi(tensors)
            all_ids = torch.cat(gathered[0])
            all_rows = torch.cat(gathered[1]).to(u_rows.dtype)
            if count_avg:
                all_counts = torch.cat(gathered[2])
        else:
            all_ids, all_rows = u_ids, u_rows
            if count_avg:
                all_counts = u_counts
        g_ids, g_rows, _ = unique_sum_rows(all_ids, all_rows)
        if count_avg:
            # total occurrence count per unique id (counts travel with the
            # rows so pre-deduplicated rows keep their multiplicity)
            _, tot, _ = unique_sum_rows(
                all_ids, all_counts.to(torch.float32).unsqueeze(1))
            g_rows = g_rows / tot.to(g_rows.dtype)
        elif self.average_sparse:
            g_rows = g_rows / float(self.comm.world_size)
        self._apply_rows(slot.info.param, g_ids, g_rows)


class SparseAllGatherEngine(_SparseEngineBase):
    """MPI-mode sparse path: raw allgather (Horovod semantics), dedup after.
    ``average_sparse`` divides by world size — hvd.allreduce(IndexedSlices,
    average=True) semantics (mpi/graph_transform.py:51-53)."""

    _average_mode = 'world'

    def reduce_and_apply(self):
        live = [s for s in self.slots if s.ids is not None]
        if self._padded and (self.comm.world_size == 1
                             or self._verify_static_counts(live)):
            for s in live:
                if self.comm.world_size == 1:
                    g_ids, g_rows, _, nvalid = unique_sum_rows_padded(
                        s.ids, s.rows)
                    if self.average_sparse:
                        pass  # /world_size == /1
                    self._apply_rows(s.info.param, g_ids, g_rows,
                                     nvalid=nvalid)
                else:
                    # raw (no local agg — Horovod semantics); the raw
                    # buffers are already fixed-shape per rank
                    self._padded_exchange_and_apply(
                        s, s.ids, s.rows, None, count_avg=False)
            return
        for s in live:
            counts = torch.ones_like(s.ids)
            self._exchange_and_apply(s, s.ids, s.rows, counts)


class SparsePSEngine(_SparseEngineBase):
    """HYBRID/PS-mode sparse path: local aggregation first (message size ∝
    unique ids), then the in-process PS exchange."""

    def __init__(self, comm, sparse_infos, optimizer, average_sparse=False,
                 local_aggregation=True, replicate_variables=True):
        super().__init__(comm, sparse_infos, optimizer, average_sparse)
        self.local_aggregation = local_aggregation
        self.sharded_params = set()
        if not replicate_variables and comm.world_size > 1:
            # shard-only storage (reference
            # ps/between_graph_parallel.py: variables live only on their
            # PS task): PartitionedEmbedding tables drop the rows they do
            # not own; lookups fetch rows over an all-to-all
            # (_ShardedLookup) and the grad push happens in its backward,
            # so reduce_and_apply for these slots is purely local.
            # Placement = byte-size greedy least-loaded across ALL tables
            # (reference GreedyLoadBalancingStrategy,
            # ps/between_graph_parallel.py:30-70), deterministic in slot
            # registration order so every rank computes the same map.
            mods, seen = [], set()
            for s in self.slots:
                mod = getattr(s.info.param, '_parallax_module', None)
                if (mod is not None and hasattr(mod, 'to_sharded')
                        and id(mod) not in seen):
                    seen.add(id(mod))
                    mods.append((mod, s.info.param))
            loads = [0] * comm.world_size
            for mod, param in mods:
                row_bytes = mod.embedding_dim * param.element_size()
                part_to_rank = []
                for size in mod.sizes:
                    r = min(range(comm.world_size),
                            key=lambda i: (loads[i], i))
                    part_to_rank.append(r)
                    loads[r] += size * row_bytes
                mod.to_sharded(comm, part_to_rank)
                self.sharded_params.add(id(param))
            if self.sharded_params:
                parallax_log.info(
                    'replicate_variables=False: %d table(s) sharded '
                    'across %d ranks', len(self.sharded_params),
                    comm.world_size)
        elif not replicate_variables:
            parallax_log.info('replicate_variables=False ignored at '
                              'world_size=1')

    def reduce_and_apply(self):
        live = [s for s in self.slots if s.ids is not None]
        # sharded tables: the cross-rank exchange already happened in the
        # lookup's backward; ids are LOCAL rows — dedup locally and apply
        for s in list(live):
            if id(s.info.param) in self.sharded_params:
                if self._padded:
                    u_ids, u_rows, _, nvalid = unique_sum_rows_padded(
                        s.ids, s.rows, average=self.average_sparse)
                    self._apply_rows(s.info.param, u_ids, u_rows,
                                     nvalid=nvalid)
                else:
                    u_ids, u_rows, _ = unique_sum_rows(
                        s.ids, s.rows, average=self.average_sparse)
                    self._apply_rows(s.info.param, u_ids, u_rows)
                live.remove(s)
        # -- sync-free padded path (fixed shapes; hipGraph-capturable at
        #    world 1; at world>1 one fixed-size allgather, no counts
        #    exchange, no padding-to-max) -------------------------------
        if self._padded and self.local_aggregation \
                and (self.comm.world_size == 1
                     or self._verify_static_counts(live)):
            pre = [_sorted_segments_presort(s.ids) for s in live]
            for s, p in zip(live, pre):
                if self.comm.world_size == 1:
                    u_ids, u_rows, _, nvalid = unique_sum_rows_padded(
                        s.ids, s.rows, average=self.average_sparse,
                        presorted=p)
                    self._apply_rows(s.info.param, u_ids, u_rows,
                                     nvalid=nvalid)
                else:
                    u_ids, u_rows, u_counts, _ = unique_sum_rows_padded(
                        s.ids, s.rows, presorted=p)
                    self._padded_exchange_and_apply(
                        s, u_ids, u_rows, u_counts,
                        count_avg=self.average_sparse)
            return
        # -- dynamic-size path ------------------------------------------
        if self.local_aggregation:
            # phase-split across tables: issue every table's sort/compare
            # first, THEN the host-syncing boundary extraction — each
            # sync then overlaps the other tables' queued GPU work
            pre = [_sorted_segments_presort(s.ids) for s in live]
        for i, s in enumerate(live):
            if self.local_aggregation:
                u_ids, u_rows, u_counts = unique_sum_rows(
                    s.ids, s.rows,
                    average=(self.average_sparse
                             and self.comm.world_size == 1),
                    presorted=pre[i])
                if self.comm.world_size == 1:
                    # already deduplicated (and averaged) — apply directly,
                    # no second sort/segment pass
                    self._apply_rows(s.info.param, u_ids, u_rows)
                    continue
            else:
                u_ids, u_rows = s.ids, s.rows
                u_counts = torch.ones_like(u_ids)
            self._exchange_and_apply(s, u_ids, u_rows, u_counts)
