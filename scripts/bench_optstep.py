"""Microbench: parallax multi_tensor_adam_dev vs torch fused/foreach Adam
on an NMT-like dense param set (~55M fp32 params in ~150 tensors).

Run on a GPU box:  python scripts/bench_optstep.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))

import torch

from parallax_amd import ops as _ops
from parallax_amd import optim as px_optim


def make_params(device):
    torch.manual_seed(0)
    shapes = []
    for _ in range(12):  # encoder-ish blocks
        shapes += [(512, 1536), (512, 512), (512,), (512,),
                   (2048, 512), (512, 2048), (2048,), (512,)]
    for _ in range(12):  # decoder-ish blocks (extra cross-attn)
        shapes += [(512, 1536), (512, 1024), (512, 512), (512, 512),
                   (512,), (512,), (2048, 512), (512, 2048), (2048,),
                   (512,)]
    shapes += [(512, 512)] * 4
    ps = [torch.nn.Parameter(torch.randn(*s, device=device)) for s in shapes]
    for p in ps:
        p.grad = torch.randn_like(p)
    n = sum(p.numel() for p in ps)
    print('%d tensors, %.1fM params' % (len(ps), n / 1e6))
    return ps


def timeit(label, fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters * 1e6
    print('%-28s %8.1f us/step' % (label, dt))
    return dt


def main():
    dev = 'cuda:0'
    ps = make_params(dev)

    opt = px_optim.Adam(ps, lr=1e-3)
    timeit('parallax Adam (dev-step)', opt.step)

    ps2 = [torch.nn.Parameter(p.detach().clone()) for p in ps]
    for p, q in zip(ps, ps2):
        q.grad = p.grad.clone()
    t_fused = torch.optim.Adam(ps2, lr=1e-3, fused=True)
    try:
        timeit('torch Adam fused', t_fused.step)
    except Exception as e:
        print('torch fused adam unavailable: %r' % e)

    ps3 = [torch.nn.Parameter(p.detach().clone()) for p in ps]
    for p, q in zip(ps, ps3):
        q.grad = p.grad.clone()
    t_foreach = torch.optim.Adam(ps3, lr=1e-3, foreach=True)
    timeit('torch Adam foreach', t_foreach.step)


if __name__ == '__main__':
    main()
