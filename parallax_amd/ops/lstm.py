"""LSTM cell with fused gate math (the LM1B hot op).

Reference analog: the LM1B model's per-step ``xw_plus_b`` [1024 -> 8192] +
sigmoid/tanh gate arithmetic + 2048->512 projection, 20 steps unrolled
(examples/lm1b/language_model.py:51-62,76-87).

MI355X-native split: the two GEMMs (input/hidden projection and the output
projection) go through torch.matmul (hipBLASLt MFMA path — a plain library
GEMM); ALL the pointwise gate math (sigmoid x3, tanh x2, the cell-state
FMA and their backward) is ONE hand-written HIP kernel pair
(ops/hip/lstm_cell.hip) so the [B, 8K] gate tensor is read/written once
instead of ~10 elementwise launches.

Gate layout in the fused gates tensor: [i | g (cell cand) | f | o] chunks of
hidden_size each, matching ``torch.chunk(gates, 4, -1)`` below.
"""

import torch
import torch.nn as nn

from parallax_amd import ops as _ops


class _LSTMGatesFn(torch.autograd.Function):
    """Fused pointwise LSTM gate fwd/bwd given pre-activation gates."""

    @staticmethod
    def forward(ctx, gates, c_prev, forget_bias):
        h, c, i_s, g_t, f_s, o_s = _ops.ext().lstm_gates_fwd(
            gates, c_prev, float(forget_bias))
        ctx.save_for_backward(c_prev, c, i_s, g_t, f_s, o_s)
        return h, c

    @staticmethod
    def backward(ctx, dh, dc):
        c_prev, c, i_s, g_t, f_s, o_s = ctx.saved_tensors
        dgates, dc_prev = _ops.ext().lstm_gates_bwd(
            dh.contiguous(), dc.contiguous() if dc is not None else
            torch.zeros_like(dh), c_prev, c, i_s, g_t, f_s, o_s)
        return dgates, dc_prev, None


def _lstm_gates_ref(gates, c_prev, forget_bias):
    i, g, f, o = torch.chunk(gates, 4, dim=-1)
    i = torch.sigmoid(i)
    g = torch.tanh(g)
    f = torch.sigmoid(f + forget_bias)
    o = torch.sigmoid(o)
    c = f * c_prev + i * g
    h = o * torch.tanh(c)
    return h, c


def lstm_gates(gates, c_prev, forget_bias=1.0):
    """Fused pointwise gate math given pre-activation gates [B, 4H]."""
    c_prev = c_prev.to(gates.dtype)
    if _ops.use_native(gates):
        return _LSTMGatesFn.apply(gates.contiguous(), c_prev.contiguous(),
                                  forget_bias)
    return _lstm_gates_ref(gates, c_prev, forget_bias)


def lstm_cell(x, h_prev, c_prev, w, b, forget_bias=1.0):
    """One LSTM step.  x [B, I], h_prev [B, H'], c_prev [B, H],
    w [(I+H'), 4H], b [4H].  Returns (h [B, H], c [B, H])."""
    xh = torch.cat([x, h_prev], dim=-1)
    gates = torch.addmm(b, xh, w)
    return lstm_gates(gates, c_prev, forget_bias)


class ParallaxLSTM(nn.Module):
    """Single-layer projected LSTM (LM1B architecture: state_size H with a
    projection H -> P applied to h, so the recurrent input is P-dim).

    Matches the reference cell: W [(I+P), 4H], projection W_P [H, P]
    (language_model.py:51-62)."""

    def __init__(self, input_size, state_size, proj_size=None,
                 forget_bias=1.0, dtype=torch.float32):
        super().__init__()
        self.input_size = input_size
        self.state_size = state_size
        self.proj_size = proj_size or state_size
        self.forget_bias = forget_bias
        in_dim = input_size + self.proj_size
        scale = in_dim ** -0.5
        self.w = nn.Parameter(
            torch.empty(in_dim, 4 * state_size, dtype=dtype).uniform_(
                -scale, scale))
        self.b = nn.Parameter(torch.zeros(4 * state_size, dtype=dtype))
        if proj_size is not None:
            self.w_proj = nn.Parameter(
                torch.empty(state_size, proj_size, dtype=dtype).uniform_(
                    -state_size ** -0.5, state_size ** -0.5))
        else:
            self.register_parameter('w_proj', None)

    def step(self, x, h_prev, c_prev):
        h, c = lstm_cell(x, h_prev, c_prev, self.w, self.b, self.forget_bias)
        if self.w_proj is not None:
            h = h @ self.w_proj
        return h, c

    def forward(self, xs, state=None):
        """xs [B, T, I] -> outputs [B, T, P], (h, c) final.

        The input half of the gate GEMM is hoisted out of the time loop:
        ONE [B*T, I] x [I, 4H] GEMM for all timesteps (big MFMA-friendly
        shape), leaving per step only the [B, P] x [P, 4H] recurrent GEMM
        + the fused gate kernel + the projection GEMM."""
        B, T, _ = xs.shape
        if state is None:
            h = xs.new_zeros(B, self.proj_size)
            c = xs.new_zeros(B, self.state_size)
        else:
            h, c = state
        w_x = self.w[:self.input_size]           # [I, 4H] (contiguous rows)
        w_h = self.w[self.input_size:]           # [P, 4H]
        gx = torch.addmm(self.b, xs.reshape(B * T, self.input_size), w_x)
        gx = gx.view(B, T, 4 * self.state_size)
        outs = []
        for t in range(T):
            gates = torch.addmm(gx[:, t], h, w_h)
            h, c = lstm_gates(gates, c, self.forget_bias)
            if self.w_proj is not None:
                h = h @ self.w_proj
            outs.append(h)
        return torch.stack(outs, dim=1), (h, c)
