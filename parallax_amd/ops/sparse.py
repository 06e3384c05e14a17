"""Indexed-slices primitives: dedup/segmented row-sum, row gather.

MI355X-native equivalent of the reference's forked
SparseConditionalAccumulator ``take_indexed_slices_grad(..., average_option)``
(contract N2; reference graph_transform_lib.py:101-102,385-390,1115-1118):
accumulate n sparse gradients by summing rows with equal indices, with an
optional per-row count average (SPARSE_AVERAGE_BY_COUNTER) vs plain sum
(SPARSE_NO_AVERAGE).

GPU path: torch.sort (rocPRIM radix sort) for ordering + hand-written HIP
scatter-atomic segment-sum kernel (ops/hip/csrc/sparse_kernels.hip).
CPU path: the same algorithm in plain PyTorch (ground truth for tests).
"""

import torch

from parallax_amd import ops as _ops

SPARSE_AVERAGE_BY_COUNTER = 1
SPARSE_NO_AVERAGE = 3


def _sorted_segments_presort(indices):
    """Device-async half of _sorted_segments: sort + boundary mask, NO
    host sync.  Several tables' presorts can be queued before any of
    their (syncing) boundary extractions run."""
    sorted_vals, order = torch.sort(indices)
    n = sorted_vals.numel()
    if n == 0:
        return sorted_vals, order, None
    mask = torch.ones(n, dtype=torch.bool, device=indices.device)
    mask[1:] = sorted_vals[1:] != sorted_vals[:-1]
    return sorted_vals, order, mask


def _sorted_segments(indices, presorted=None):
    """Sort indices; return (sorted_vals, order, seg_starts, unique_vals).

    seg_starts has one entry per unique index (start position in the sorted
    array) plus a final n sentinel.  The ``torch.nonzero`` here is the one
    host sync of the pipeline (dynamic unique count)."""
    sorted_vals, order, mask = (presorted if presorted is not None
                                else _sorted_segments_presort(indices))
    n = sorted_vals.numel()
    if n == 0:
        empty = torch.zeros(1, dtype=torch.long, device=indices.device)
        return sorted_vals, order, empty, sorted_vals
    seg_starts = torch.nonzero(mask).flatten()
    unique_vals = sorted_vals[seg_starts]
    seg_starts = torch.cat([seg_starts,
                            torch.tensor([n], dtype=torch.long,
                                         device=indices.device)])
    return sorted_vals, order, seg_starts, unique_vals


def unique_sum_rows(indices, values, average=False, presorted=None):
    """Deduplicate ``indices`` and segment-sum the corresponding rows of
    ``values`` ([n, d] or [n]); returns (unique_indices[u], summed[u, d],
    counts[u]).  ``average=True`` divides each output row by its count
    (reference average_option=SPARSE_AVERAGE_BY_COUNTER); ``presorted``
    takes a _sorted_segments_presort result to pipeline several tables."""
    assert indices.dim() == 1
    squeeze = False
    if values.dim() == 1:
        values = values.unsqueeze(1)
        squeeze = True
    assert values.shape[0] == indices.shape[0]

    if indices.numel() == 0:
        out = values.new_zeros((0, values.shape[1]))
        counts = torch.zeros(0, dtype=torch.long, device=indices.device)
        return indices, out.squeeze(1) if squeeze else out, counts

    sorted_vals, order, seg_starts, unique_vals = \
        _sorted_segments(indices, presorted)
    counts = seg_starts[1:] - seg_starts[:-1]
    if _ops.use_native(values):
        out = _ops.ext().segment_sum_rows(
            values, order.to(torch.int32), seg_starts.to(torch.int32),
            bool(average))
    else:
        out = segment_sum_rows_ref(values, order, seg_starts, average)
    return unique_vals, (out.squeeze(1) if squeeze else out), counts


def unique_sum_rows_padded(indices, values, average=False, presorted=None,
                           in_counts=None):
    """Sync-free fixed-shape dedup: returns (ids[n], rows[n, d],
    counts[n], nvalid) where n == indices.numel() and only the first
    ``nvalid`` (a 0-dim DEVICE tensor) entries are real; the tail is
    (id 0, zero rows, zero counts).  Zero-rows applied to row 0 are
    no-ops for sum-style row updates, and consumers that cannot tolerate
    them mask with ``nvalid`` (the native sparse optimizer kernels do).

    Unlike unique_sum_rows there is NO host synchronization (no dynamic
    unique count), so the whole sparse tail is hipGraph-capturable and
    costs no ~70 us/table stall (round-1 TODO item 3).

    ``in_counts``: pre-aggregated per-entry multiplicities (float) whose
    segment SUMS become the output counts (the second dedup of a padded
    exchange); default = occurrence counting.
    """
    assert indices.dim() == 1
    squeeze = False
    if values.dim() == 1:
        values = values.unsqueeze(1)
        squeeze = True
    n = indices.numel()
    dev = indices.device
    if n == 0:
        out = values.new_zeros((0, values.shape[1]))
        counts = torch.zeros(0, dtype=torch.float32, device=dev)
        nvalid = torch.zeros((), dtype=torch.long, device=dev)
        return indices, out.squeeze(1) if squeeze else out, counts, nvalid
    sorted_vals, order, mask = (presorted if presorted is not None
                                else _sorted_segments_presort(indices))
    seg_id = torch.cumsum(mask, 0) - 1          # segment of each sorted pos
    nvalid = seg_id[-1] + 1
    uniq = torch.zeros(n, dtype=indices.dtype, device=dev)
    uniq.scatter_(0, seg_id, sorted_vals)
    gathered = values.index_select(0, order).to(torch.float32)
    out = torch.zeros((n, values.shape[1]), dtype=torch.float32, device=dev)
    out.index_add_(0, seg_id, gathered)
    counts = torch.zeros(n, dtype=torch.float32, device=dev)
    if in_counts is None:
        counts.scatter_add_(0, seg_id,
                            torch.ones(n, dtype=torch.float32, device=dev))
    else:
        counts.scatter_add_(0, seg_id,
                            in_counts.index_select(0, order)
                            .to(torch.float32))
    if average:
        out = out / counts.clamp(min=1.0).unsqueeze(1)
    out = out.to(values.dtype)
    return uniq, (out.squeeze(1) if squeeze else out), counts, nvalid


def segment_sum_rows(values, order, seg_starts, average=False):
    """Row segment-sum given precomputed sort ``order`` and segment
    boundaries (see _sorted_segments).  out[s] = sum_k values[order[k]]
    for k in [seg_starts[s], seg_starts[s+1])."""
    if _ops.use_native(values):
        return _ops.ext().segment_sum_rows(
            values, order.to(torch.int32), seg_starts.to(torch.int32),
            bool(average))
    return segment_sum_rows_ref(values, order, seg_starts, average)


def segment_sum_rows_ref(values, order, seg_starts, average=False):
    """CPU/ground-truth implementation (fp32 accumulation)."""
    u = seg_starts.numel() - 1
    gathered = values.index_select(0, order).to(torch.float32)
    out = torch.zeros((u, values.shape[1]), dtype=torch.float32,
                      device=values.device)
    seg_ids = torch.repeat_interleave(
        torch.arange(u, device=values.device),
        (seg_starts[1:] - seg_starts[:-1]))
    out.index_add_(0, seg_ids, gathered)
    if average:
        counts = (seg_starts[1:] - seg_starts[:-1]).to(torch.float32)
        out = out / counts.unsqueeze(1)
    return out.to(values.dtype)


def rows_gather(weight, indices):
    """out[i] = weight[indices[i]] — the embedding-lookup gather
    (hand-written HIP kernel on GPU; reference: TF Gather CUDA kernel,
    examples/lm1b/language_model.py:26-40)."""
    if _ops.use_native(weight):
        return _ops.ext().rows_gather(weight, indices.to(torch.int64))
    return weight.index_select(0, indices.reshape(-1)).view(
        *indices.shape, weight.shape[1])
