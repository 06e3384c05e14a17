"""Correctness + TFLOPS of the hand-written MFMA GEMM vs hipBLASLt
(torch.matmul), bf16, A @ B^T."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from parallax_amd import ops

e = ops.ext()
dev = 'cuda:0'
torch.manual_seed(0)


def check(M, N, K):
    a = (torch.rand(M, K, device=dev, dtype=torch.bfloat16) * 2 - 1)
    b = (torch.rand(N, K, device=dev, dtype=torch.bfloat16) * 2 - 1)
    c = e.pxgemm_bt(a, b)
    ref = (a.float() @ b.float().t())
    err = (c.float() - ref).abs().max().item()
    rel = err / ref.abs().max().clamp_min(1e-6).item()
    # identity-A transpose check with ASYMMETRIC B (guide G9)
    ai = torch.zeros(M, K, device=dev, dtype=torch.bfloat16)
    d = min(M, K)
    ai[torch.arange(d), torch.arange(d)] = 1.0
    ci = e.pxgemm_bt(ai, b)
    ref_i = ai.float() @ b.float().t()
    ok_t = torch.allclose(ci.float(), ref_i, atol=1e-2)
    rel8 = 0.0
    ok8 = True
    if hasattr(e, 'pxgemm8_bt'):
        c8 = e.pxgemm8_bt(a, b)
        rel8 = float((c8.float() - ref).abs().max() /
                     ref.abs().max().clamp_min(1e-6))
        ok8 = torch.allclose(e.pxgemm8_bt(ai, b).float(), ref_i, atol=1e-2)
    print('M=%5d N=%5d K=%5d  2ph rel=%.5f idA=%s   8ph rel=%.5f idA=%s'
          % (M, N, K, rel, ok_t, rel8, ok8), flush=True)
    return rel < 0.02 and ok_t and rel8 < 0.02 and ok8


def bench(M, N, K, iters=30):
    a = (torch.rand(M, K, device=dev, dtype=torch.bfloat16) * 2 - 1)
    b = (torch.rand(N, K, device=dev, dtype=torch.bfloat16) * 2 - 1)

    def t(fn):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    t_ours = t(lambda: e.pxgemm_bt(a, b))
    t_p8 = t(lambda: e.pxgemm8_bt(a, b)) if hasattr(e, 'pxgemm8_bt') else 0
    t_blas = t(lambda: a @ b.t())
    fl = 2.0 * M * N * K
    print('M=%5d N=%5d K=%5d  2ph %7.1f us (%6.1f TF)  8ph %7.1f us '
          '(%6.1f TF)  hipblaslt %7.1f us (%6.1f TF)'
          % (M, N, K, t_ours * 1e6, fl / t_ours / 1e12,
             t_p8 * 1e6, fl / max(t_p8, 1e-12) / 1e12,
             t_blas * 1e6, fl / t_blas / 1e12), flush=True)


if __name__ == '__main__':
    ok = True
    for shape in ((128, 128, 64), (256, 256, 128), (512, 384, 192),
                  (100, 70, 64), (3136, 512, 2048),  # edge tiles
                  (2560, 8192, 512), (4096, 4096, 4096)):
        ok &= check(*shape)
    print('CORRECTNESS:', 'PASS' if ok else 'FAIL', flush=True)
    for shape in ((4096, 4096, 4096), (8192, 8192, 8192),
                  (2560, 8192, 512),   # LM1B gx GEMM
                  (128, 8192, 512),    # LM1B recurrent gates (fwd)
                  (128, 512, 8192),    # LM1B bwd carry dG @ w_h^T
                  (128, 2048, 512),    # LM1B bwd ds
                  (128, 512, 2048),    # LM1B proj fwd
                  (200704, 256, 64),   # ResNet 1x1 56x56
                  (50176, 512, 128), (12544, 1024, 256)):
        try:
            bench(*shape)
        except Exception as e:
            print('shape', shape, 'skipped:', e, flush=True)
