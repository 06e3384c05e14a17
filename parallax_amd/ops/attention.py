"""Fused short-sequence attention (ops/hip/csrc/attention.hip).

One 256-thread workgroup per (batch, head): q/k/v tiles, scores, softmax
and the P @ V product all live in LDS; backward recomputes P.  Eligible
when T, S <= 64 and d_head == 64 (the NMT transformer's 48x48x64 shape,
where library flash-attention kernels spend 17/28 us on ~2 us of HBM
traffic); everything else falls back to torch SDPA.

Measured on MI355X: isolated wall-clock is at parity (fwd+bwd 83.7 vs
83.8 us, scripts/bench_attn.py) but that comparison flatters us — the
torch SDPA *call* carries ~15 us of dispatch overhead that disappears
under hipGraph capture.  In-chain KERNEL time (rocprofv3, captured NMT
step): ours 28.1 us fwd / 63.9 us bwd vs aotriton's 16.9 / 28.2 — ours
loses ~1.0 ms/step e2e (308k -> 271k words/s).  Root cause: the
per-(t,s) dot products re-read both operand rows from LDS for every
MAC (~0.6 MB of LDS traffic per workgroup fwd, 3x that bwd, against a
~65 TB/s aggregate LDS ceiling); aotriton's MFMA tiles amortize
operands in registers.  The fix is a register-/MFMA-tiled rewrite
(TODO.md), not a split.  DEFAULT-ON since the round-2 MFMA rewrite
(fwd 6.8 us / bwd 19.3 us vs SDPA-aotriton 33/63 at the NMT shape —
gpurun_out/attn_ab_mfma.log); PARALLAX_PX_ATTN=0 restores SDPA; was
torch SDPA.
"""

import os

import torch
import torch.nn.functional as F

from parallax_amd import ops as _ops


class _PxAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal):
        ctx.save_for_backward(q, k, v)
        ctx.causal = causal
        return _ops.ext().px_attn_fwd(q, k, v, causal)

    @staticmethod
    def backward(ctx, dout):
        q, k, v = ctx.saved_tensors
        dq, dk, dv = _ops.ext().px_attn_bwd(q, k, v, dout.contiguous(),
                                            ctx.causal)
        return dq, dk, dv, None


def _eligible(q, k, v):
    return (q.is_cuda and q.dtype == torch.bfloat16
            and k.dtype == torch.bfloat16 and v.dtype == torch.bfloat16
            and q.dim() == 4 and q.shape[-1] == 64
            and q.shape[2] <= 64 and k.shape[2] <= 64
            and q.stride(3) == 1 and k.stride(3) == 1 and v.stride(3) == 1
            and all(t.stride(i) % 4 == 0
                    for t in (q, k, v) for i in (0, 1, 2))
            and _ops.native_available()
            and os.environ.get('PARALLAX_PX_ATTN', '1') == '1')


def px_attention(q, k, v, is_causal=False):
    """[B, H, T/S, D] q/k/v -> [B, H, T, D]; torch-SDPA-compatible."""
    if _eligible(q, k, v):
        return _PxAttnFn.apply(q, k, v, is_causal)
    return F.scaled_dot_product_attention(q, k, v, is_causal=is_causal)
