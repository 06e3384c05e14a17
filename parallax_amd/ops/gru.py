"""Fused GRU gate math (skip_thoughts hot op) — HIP kernel on GPU,
torch composition on CPU."""

import torch

from parallax_amd import ops as _ops


class _GRUGatesFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, xg, hg, h_prev):
        h_out, r_s, z_s, n_t, hgn_s = _ops.ext().gru_gates_fwd(
            xg, hg, h_prev)
        ctx.save_for_backward(h_prev, r_s, z_s, n_t, hgn_s)
        return h_out

    @staticmethod
    def backward(ctx, dh):
        h_prev, r_s, z_s, n_t, hgn_s = ctx.saved_tensors
        d_xg, d_hg, d_hprev = _ops.ext().gru_gates_bwd(
            dh, h_prev, r_s, z_s, n_t, hgn_s)
        return d_xg, d_hg, d_hprev


def gru_gates(xg, hg, h_prev):
    """xg = x@W_x + b [B, 3H] (layout r|z|n), hg = h@W_h [B, 3H]:
    r = sig(xg_r + hg_r); z = sig(xg_z + hg_z);
    n = tanh(xg_n + r * hg_n); h' = (1-z)*n + z*h_prev."""
    if _ops.use_native(xg):
        return _GRUGatesFn.apply(xg.contiguous(),
                                 hg.to(xg.dtype).contiguous(),
                                 h_prev.to(xg.dtype).contiguous())
    H = xg.shape[1] // 3
    r = torch.sigmoid(xg[:, :H] + hg[:, :H])
    z = torch.sigmoid(xg[:, H:2 * H] + hg[:, H:2 * H])
    n = torch.tanh(xg[:, 2 * H:] + r * hg[:, 2 * H:])
    return (1 - z) * n + z * h_prev
