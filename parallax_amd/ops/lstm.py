"""LSTM cell with fused gate math (the LM1B hot op).

Reference analog: the LM1B model's per-step ``xw_plus_b`` [1024 -> 8192] +
sigmoid/tanh gate arithmetic + 2048->512 projection, 20 steps unrolled
(examples/lm1b/language_model.py:51-62,76-87).

MI355X-native split: the GEMMs (input projection hoisted out of the time
loop, the recurrent/projection GEMMs, the batched weight-grad GEMMs) run
on hipBLASLt by default (the hand-written pxgemm substitution is opt-in,
see _LSTMSeqFn); ALL the pointwise gate math (sigmoid x3, tanh x2, the
cell-state FMA and their backward) is ONE hand-written HIP kernel pair
(ops/hip/lstm_cell.hip) so the [B, 8K] gate tensor is read/written once
instead of ~10 elementwise launches.  ParallaxLSTM.forward dispatches the
whole unroll to the sequence-level Function (_LSTMSeqFn) on GPU.

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


class _LSTMSeqFn(torch.autograd.Function):
    """Sequence-level projected-LSTM autograd: the whole T-step unroll as
    ONE Function so backward can (a) run 3 launches per step (fused gate
    kernel + 2 GEMMs) instead of autograd's ~8, and (b) batch BOTH weight
    gradients into single [K, B*T] x [B*T, N] GEMMs at the end — the
    per-step dW accumulation GEMMs of eager autograd disappear.

    Inputs: gx [B,T,4H] (pre-computed input projections + bias), h0 [B,P],
    c0 [B,H], w_h [P,4H], w_proj [H,P].  Outputs: hp [B,T,P], c_T.
    h0/c0 receive no grads (recurrent state is detached between steps —
    truncated BPTT, reference language_model_graph.py:76-87).
    """

    @staticmethod
    def forward(ctx, gx, h0, c0, w_h, w_proj, forget_bias):
        ext = _ops.ext()
        B, T, G = gx.shape
        H = G // 4
        P = w_proj.shape[1]
        # one compute-dtype cast per sequence (autocast would cast per use)
        ctx.w_dtype = w_h.dtype
        w_h = w_h.to(gx.dtype)
        w_proj = w_proj.to(gx.dtype)
        # pre-allocated stacks; every per-step kernel/GEMM writes straight
        # into its slice (out-variant gate kernels + mm(out=...)) — the
        # time loop issues NO copy kernels
        hp_all = torch.empty(T + 1, B, P, dtype=gx.dtype, device=gx.device)
        s_all = torch.empty(T, B, H, dtype=gx.dtype, device=gx.device)
        c_all = torch.empty(T + 1, B, H, dtype=gx.dtype, device=gx.device)
        acts = torch.empty(4, T, B, H, dtype=torch.float32,
                           device=gx.device)
        c_all[0] = c0.to(gx.dtype)
        hp_all[0] = h0.to(gx.dtype)
        import os as _os
        # round-2 default: the fused step kernel (gates GEMM + pointwise
        # in ONE launch, gate-grouped columns so the pointwise is
        # register-local in the MFMA C fragments) — the [B,4H] gates
        # tensor never exists in memory and the dependent chain loses a
        # launch per step (lstm_fused.hip)
        use_fused = (gx.dtype == torch.bfloat16 and B % 64 == 0
                     and B <= 128 and H % 32 == 0 and P % 64 == 0
                     and hasattr(ext, 'lstm_step_fwd_fused')
                     and _os.environ.get('PARALLAX_FUSED_LSTM',
                                         '1') != '0')
        # pxgemm beats hipBLASLt 2.4x on the recurrent gates shape in
        # ISOLATION (7.7 vs 18.8 us) but measured ~4%% SLOWER end-to-end
        # inside the captured step (A/B fwd=0/bwd=0 756.7k vs fwd=1/bwd=1
        # 727.4k words/s) — dependent-chain pipelining favors the library
        # kernels here, so the substitution is opt-in
        # (PARALLAX_PX_LSTM_FWD/BWD=1)
        use_px = (not use_fused and gx.dtype == torch.bfloat16
                  and B % 128 == 0
                  and (4 * H) % 128 == 0 and P % 64 == 0
                  and _os.environ.get('PARALLAX_PX_LSTM_FWD', '0') == '1')
        w_hT = w_h.t().contiguous() if (use_px or use_fused) else None
        ctx.used_fused = use_fused
        # per-timestep projection on the split-K skinny GEMM (the chain's
        # dependent [B,P] = [B,H] @ [H,P] launches: hipBLASLt ~19 us,
        # px_skinny ~5 us + cast)
        use_skinny = (gx.dtype == torch.bfloat16 and B <= 128
                      and P % 64 == 0 and H % 64 == 0
                      and hasattr(ext, 'px_skinny_bt')
                      and _os.environ.get('PARALLAX_SKINNY', '1') == '1')
        if use_skinny:
            w_projT = w_proj.t().contiguous()
            ws_p = torch.zeros(B, P, dtype=torch.float32,
                               device=gx.device)
            cnt_p = torch.zeros(P // 64, dtype=torch.int32,
                                device=gx.device)
        for t in range(T):
            if use_fused:
                ext.lstm_step_fwd_fused(
                    hp_all[t], w_hT, gx[:, t], c_all[t],
                    float(forget_bias), s_all[t], c_all[t + 1],
                    acts[0, t], acts[1, t], acts[2, t], acts[3, t])
            else:
                if use_px:
                    gates = ext.pxgemm_bt_add(hp_all[t], w_hT, gx[:, t])
                else:
                    gates = torch.addmm(gx[:, t], hp_all[t], w_h)
                ext.lstm_gates_fwd_out(gates, c_all[t], float(forget_bias),
                                       s_all[t], c_all[t + 1], acts[0, t],
                                       acts[1, t], acts[2, t], acts[3, t])
            if use_skinny:
                ext.px_skinny_bt(s_all[t], w_projT, ws_p, cnt_p,
                                 hp_all[t + 1], None)
            else:
                torch.mm(s_all[t], w_proj, out=hp_all[t + 1])
        ctx.save_for_backward(hp_all, s_all, c_all, acts, w_h, w_proj)
        ctx.fbias = forget_bias
        # outs view [B, T, P]: non-contiguous transpose of hp_all[1:]; the
        # consumer (reshape for the softmax GEMM) makes it contiguous once
        return hp_all[1:].transpose(0, 1), c_all[T].clone()

    @staticmethod
    def backward(ctx, d_outs, d_cT):
        ext = _ops.ext()
        hp_all, s_all, c_all, acts, w_h, w_proj = ctx.saved_tensors
        T, B, H = s_all.shape
        P = hp_all.shape[2]
        dG = torch.empty(T, B, 4 * H, dtype=w_h.dtype, device=w_h.device)
        dHP = torch.empty(T, B, P, dtype=w_h.dtype, device=w_h.device)
        ds = torch.empty(B, H, dtype=w_h.dtype, device=w_h.device)
        dc_a = torch.empty(B, H, dtype=w_h.dtype, device=w_h.device)
        dc_b = torch.empty(B, H, dtype=w_h.dtype, device=w_h.device)
        w_h_t = w_h.t().contiguous()
        if d_cT is None:
            dc_a.zero_()
        else:
            dc_a.copy_(d_cT)
        dc, dc_next = dc_a, dc_b
        import os as _os
        # fused bwd step: ds GEMM + gate backward in one launch
        # (w_proj's natural [H, P] layout IS the transposed operand)
        use_fused = (getattr(ctx, 'used_fused', False)
                     and hasattr(ext, 'lstm_step_bwd_fused')
                     and _os.environ.get('PARALLAX_FUSED_LSTM',
                                         '1') != '0')
        # ds = dhp @ w_proj^T maps DIRECTLY onto pxgemm_bt (w_proj is
        # [H, P] row-major = the transposed operand): 7.6 vs 19.3 us
        use_px = (not use_fused and w_h.dtype == torch.bfloat16
                  and B % 128 == 0
                  and H % 128 == 0 and P % 64 == 0
                  and _os.environ.get('PARALLAX_PX_LSTM_BWD', '0') == '1')
        w_proj_t = None if use_fused else w_proj.t().contiguous()
        # the recurrent carry dhp = dG @ w_h^T + upstream slice on the
        # split-K skinny GEMM with the add fused into its cast epilogue
        use_skinny = (w_h.dtype == torch.bfloat16
                      and d_outs.dtype == torch.bfloat16 and B <= 128
                      and P % 64 == 0 and (4 * H) % 64 == 0
                      and d_outs.stride(2) == 1
                      and hasattr(ext, 'px_skinny_bt')
                      and _os.environ.get('PARALLAX_SKINNY', '1') == '1')
        if use_skinny:
            ws_c = torch.zeros(B, P, dtype=torch.float32,
                               device=w_h.device)
            cnt_c = torch.zeros(P // 64, dtype=torch.int32,
                                device=w_h.device)
        for t in range(T - 1, -1, -1):
            if t == T - 1:
                dHP[t].copy_(d_outs[:, t])
            elif use_skinny:
                ext.px_skinny_bt(dG[t + 1], w_h, ws_c, cnt_c, dHP[t],
                                 d_outs[:, t])
            else:
                # dhp = upstream slice + recurrent carry; NOTE an addmm
                # (beta=1) fusion of the add measured ~2% SLOWER overall
                # (hipBLASLt picks a worse kernel on the strided-C path)
                torch.add(d_outs[:, t].to(w_h.dtype),
                          torch.mm(dG[t + 1], w_h_t), out=dHP[t])
            if use_fused:
                ext.lstm_step_bwd_fused(
                    dHP[t], w_proj, dc, c_all[t], c_all[t + 1],
                    acts[0, t], acts[1, t], acts[2, t], acts[3, t],
                    dG[t], dc_next)
            else:
                if use_px:
                    ds = ext.pxgemm_bt(dHP[t], w_proj)
                else:
                    torch.mm(dHP[t], w_proj_t, out=ds)
                ext.lstm_gates_bwd_out(ds, dc, c_all[t], c_all[t + 1],
                                       acts[0, t], acts[1, t], acts[2, t],
                                       acts[3, t], dG[t], dc_next)
            dc, dc_next = dc_next, dc
        # batched weight grads: one GEMM each over all timesteps
        dG_flat = dG.reshape(T * B, 4 * H)
        dw_h = hp_all[:T].reshape(T * B, P).t() @ dG_flat
        dw_proj = s_all.reshape(T * B, H).t() @ dHP.reshape(T * B, P)
        dgx = dG.permute(1, 0, 2)
        return dgx, None, None, dw_h.to(ctx.w_dtype), \
            dw_proj.to(ctx.w_dtype), None


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
        if (self.w_proj is not None and _ops.native_available()
                and gx.is_cuda):
            outs, c = _LSTMSeqFn.apply(gx, h, c, w_h, self.w_proj,
                                       self.forget_bias)
            return outs, (outs[:, -1], c)
        outs = []
        for t in range(T):
            gates = torch.addmm(gx[:, t], h, w_h)
            h, c = lstm_gates(gates, c, self.forget_bias)
            if self.w_proj is not None:
                h = h @ self.w_proj
            outs.append(h)
        return torch.stack(outs, dim=1), (h, c)
