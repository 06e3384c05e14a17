"""Microbench the sparse-reduce pieces at LM1B shapes."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from parallax_amd import ops
from parallax_amd.ops.sparse import unique_sum_rows, _sorted_segments

e = ops.ext()
dev = 'cuda:0'
torch.manual_seed(0)


def timeit(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


for tag, n, d, vmax in (('emb', 2560, 512, 793470),
                        ('softmax_w', 10752, 512, 793470),
                        ('bias', 10752, 1, 793470)):
    ids = torch.randint(0, vmax, (n,), device=dev)
    # realistic: sampled ids are Zipfian -> heavy duplication of small ids
    from parallax_amd.ops.sampled_softmax import log_uniform_candidates
    if n > 2560:
        zipf, _ = log_uniform_candidates(8192, vmax, dev)
        ids = torch.cat([ids[:n - 8192], zipf])
    vals = torch.randn(n, d, device=dev)
    sorted_vals, order, seg_starts, uniq = _sorted_segments(ids)
    u = uniq.numel()
    order32 = order.to(torch.int32)
    seg32 = seg_starts.to(torch.int32)

    t_sort = timeit(lambda: torch.sort(ids))
    t_seg = timeit(lambda: e.segment_sum_rows(vals, order32, seg32, False))
    t_all = timeit(lambda: unique_sum_rows(ids, vals))
    print('%10s n=%5d u=%5d d=%4d  sort=%7.1fus  segkernel=%7.1fus  '
          'unique_sum_total=%7.1fus' % (tag, n, u, d, t_sort, t_seg, t_all))

# the apply kernel
w = torch.randn(793470, 512, device=dev)
acc = torch.ones(793470, 512, device=dev)
ids_u = torch.unique(torch.randint(0, 793470, (10752,), device=dev))
rows = torch.randn(ids_u.numel(), 512, device=dev)
t_ad = timeit(lambda: e.sparse_adagrad(w, acc, ids_u, rows, 0.1, 1e-10))
print('sparse_adagrad u=%d: %7.1fus' % (ids_u.numel(), t_ad))
