"""pxgemm_bt vs hipBLASLt on the NMT transformer's GEMM shapes
(M = batch*seq = 3072, d_model 512, ffn 2048; fwd + bwd orientations)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from parallax_amd import ops

e = ops.ext()
dev = 'cuda:0'
torch.manual_seed(0)


def t(fn, iters=50):
    for _ in range(8):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench(M, N, K, tag):
    a = (torch.rand(M, K, device=dev, dtype=torch.bfloat16) * 2 - 1)
    b = (torch.rand(N, K, device=dev, dtype=torch.bfloat16) * 2 - 1)
    c = e.pxgemm_bt(a, b)
    ref = a.float() @ b.float().t()
    rel = (c.float() - ref).abs().max().item() / \
        ref.abs().max().clamp_min(1e-6).item()
    t_ours = t(lambda: e.pxgemm_bt(a, b))
    t_blas = t(lambda: a @ b.t())
    fl = 2.0 * M * N * K
    print('%-22s M=%6d N=%5d K=%5d ours %6.1fus (%5.1fTF) blas %6.1fus '
          '(%5.1fTF) rel=%.4f' % (tag, M, N, K, t_ours * 1e6,
                                  fl / t_ours / 1e12, t_blas * 1e6,
                                  fl / t_blas / 1e12, rel), flush=True)


if __name__ == '__main__':
    B = 3072
    for (M, N, K), tag in [
        ((B, 1536, 512), 'qkv fwd'),
        ((B, 512, 512), 'attn out fwd'),
        ((B, 2048, 512), 'fc1 fwd'),
        ((B, 512, 2048), 'fc2 fwd'),
        ((B, 512, 1536), 'qkv bwd dx'),
        ((1536, 512, B), 'qkv bwd dw'),
        ((B, 2048, 512), 'fc2 bwd dx'),
        ((2048, 512, B), 'fc1 bwd dw'),
        ((B, 4097, 512), 'sampled logits'),
    ]:
        bench(M, N, K, tag)
