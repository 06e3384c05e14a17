"""Fused short-sequence attention (ops/hip/csrc/attention.hip).

One 256-thread (4-wave) workgroup per (batch, head); eligible when
T, S <= 64 and d_head == 64 (the NMT transformer's 48x48x64 shape);
everything else falls back to torch SDPA.

DEFAULT-ON since the round-2 MFMA rewrite: every GEMM runs on
mfma_f32_16x16x32_bf16 over zero-padded 64x64 LDS tiles (reductions
over the non-contiguous axis read transposed twin images written at
load; softmax row-reduces are 4-step __shfl_xor inside the C-fragment
lane groups; the backward's dS images alias dead tiles to keep 2
workgroups/CU).  Outputs are allocated [B,T,H,D]-permuted so the
model's transpose+reshape afterwards is a free view.  Measured
(MI355X, scripts/bench_attn.py): fwd 6.8 us / bwd 12.2 us vs torch
SDPA (aotriton) 33 / 63; fwd+bwd 19.8 vs 104.6 us; NMT e2e ahead of
the SDPA path.  PARALLAX_PX_ATTN=0 restores SDPA.

History: the round-1 scalar-VALU version of these kernels lost
in-chain (28/64 us — every MAC re-read both operand rows from LDS);
its analysis lives in git and TODO.md item 6b.
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
    # causal with T != S is convention-ambiguous (top-left vs
    # bottom-right alignment); the kernel implements the T == S case the
    # models use — route anything else to SDPA
    if _eligible(q, k, v) and not (is_causal and q.shape[2] != k.shape[2]):
        return _PxAttnFn.apply(q, k, v, is_causal)
    return F.scaled_dot_product_attention(q, k, v, is_causal=is_causal)
