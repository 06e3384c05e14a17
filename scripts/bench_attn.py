"""Isolated px_attn vs torch SDPA (aotriton) timing on the NMT shape."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import torch.nn.functional as F
from parallax_amd import ops
e = ops.ext()
B, H, T, D = 64, 8, 48, 64
q = torch.randn(B, H, T, D, device='cuda', dtype=torch.bfloat16)
k, v = torch.randn_like(q), torch.randn_like(q)
do = torch.randn_like(q)


def t(fn, iters=100):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


print('px fwd   %7.1f us' % t(lambda: e.px_attn_fwd(q, k, v, True)))
print('sdpa fwd %7.1f us' % t(lambda: F.scaled_dot_product_attention(
    q, k, v, is_causal=True)))
print('px bwd   %7.1f us' % t(lambda: e.px_attn_bwd(q, k, v, do, True)))
qg = q.clone().requires_grad_(True)


def sdpa_full():
    o = F.scaled_dot_product_attention(qg, k, v, is_causal=True)
    o.backward(do)
    qg.grad = None


print('sdpa f+b %7.1f us' % t(sdpa_full))
print('px f+b   %7.1f us' % t(lambda: (e.px_attn_fwd(q, k, v, True),
                                       e.px_attn_bwd(q, k, v, do, True))))
