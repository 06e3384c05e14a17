"""Within-probe interleaved A/B of pxgemm8 schedule variants vs
hipBLASLt (guide §5.4 rule 24: N variants x M rounds in ONE process,
report median + min)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from parallax_amd import ops

e = ops.ext()
dev = 'cuda:0'
torch.manual_seed(0)


def run(M, N, K, rounds=9, iters=20):
    a = torch.rand(M, K, device=dev, dtype=torch.bfloat16) * 2 - 1
    b = torch.rand(N, K, device=dev, dtype=torch.bfloat16) * 2 - 1
    fl = 2.0 * M * N * K
    variants = {
        'v0-8ph': lambda: e.pxgemm8_bt_v(a, b, 0),
        'v1-merged': lambda: e.pxgemm8_bt_v(a, b, 1),
        'v2-pipe': lambda: e.pxgemm8_bt_v(a, b, 2),
        'v3-1bar': lambda: e.pxgemm8_bt_v(a, b, 3),
        'tall': lambda: e.px_tall_bt(a, b),
        'hipblaslt': lambda: a @ b.t(),
    }
    # correctness first
    ref = a.float() @ b.float().t()
    for name, fn in variants.items():
        c = fn()
        rel = float((c.float() - ref).abs().max() / ref.abs().max())
        assert rel < 0.02, (name, rel)
    times = {k: [] for k in variants}
    for name, fn in variants.items():  # warm
        for _ in range(5):
            fn()
    torch.cuda.synchronize()
    for _ in range(rounds):
        for name, fn in variants.items():
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(iters):
                fn()
            torch.cuda.synchronize()
            times[name].append((time.perf_counter() - t0) / iters)
    print('M=%d N=%d K=%d' % (M, N, K))
    for name, ts in times.items():
        ts = sorted(ts)
        med = ts[len(ts) // 2]
        print('  %-10s median %8.1f us (%7.1f TF)   min %8.1f us (%7.1f TF)'
              % (name, med * 1e6, fl / med / 1e12,
                 ts[0] * 1e6, fl / ts[0] / 1e12), flush=True)


if __name__ == '__main__':
    shapes = [(4096, 4096, 4096), (8192, 8192, 8192), (2560, 8192, 512),
              (12544, 1024, 256), (3136, 512, 4608)]
    if len(sys.argv) > 1:
        shapes = [tuple(int(x) for x in s.split(',')) for s in sys.argv[1:]]
    for s in shapes:
        run(*s)
