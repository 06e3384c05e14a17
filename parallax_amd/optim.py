"""Fused optimizers — the reference's update-op vocabulary, MI355X-native.

The reference's graph rewrites recognize a fixed set of variable-update ops
(graph_transform_lib.py:56-75): dense {ApplyGradientDescent, ApplyMomentum,
ApplyAdagrad, ApplyAdam, ApplyRMSProp, ...} and sparse {SparseApplyAdagrad,
ScatterAdd/Sub, ...}.  Here each optimizer provides

  * a fused DENSE step — one hand-written multi-tensor HIP launch over all
    its parameters (ops/hip/multi_tensor.hip) on GPU, a torch reference on
    CPU; and
  * ``sparse_row_step(param, ids, grad_rows, counts)`` — the
    SparseApplyAdagrad-equivalent: update ONLY the touched (deduplicated)
    rows, with per-row optimizer state.  The sparse engines call this on
    the owner rank of each shard.

All sparse state (e.g. the Adagrad accumulator) is kept per-parameter at
full shape so checkpoints can round-trip it.
"""

import torch
from torch.optim.optimizer import Optimizer

from parallax_amd import ops as _ops


def _use_native(params):
    return any(p.is_cuda for p in params) and _ops.native_available()


def _dtype_groups(params, *lists):
    """Split aligned per-param lists into same-dtype groups: the
    multi-tensor kernels dispatch ONE element type per launch (a mixed
    bf16-conv + fp32-BN model would otherwise reinterpret memory)."""
    dtypes = {p.dtype for p in params}
    if len(dtypes) == 1:
        yield (params,) + lists
        return
    order = []
    by_dt = {}
    for i, p in enumerate(params):
        if p.dtype not in by_dt:
            by_dt[p.dtype] = []
            order.append(p.dtype)
        by_dt[p.dtype].append(i)
    for dt in order:
        ii = by_dt[dt]
        yield tuple([lst[i] for i in ii] for lst in (params,) + lists)


class ParallaxOptimizer(Optimizer):
    """Base: splits the fused dense path from the per-row sparse path."""

    def supports_sparse(self):
        return True

    def supports_padded_sparse(self):
        """True when sparse_row_step accepts fixed-shape padded inputs
        (tail = id 0 / zero rows, bounded by a DEVICE nvalid scalar) —
        the sync-free sparse path.  Requires either pure sum-style index
        updates (safe under zero-duplicates) or an nvalid-masked native
        kernel; Adam/RMSProp's index_copy_ row state is not safe under
        duplicate tail ids, so they opt out."""
        return False

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            params, grads = [], []
            for p in group['params']:
                if p.grad is None:
                    continue
                if getattr(p.grad, '_parallax_swallowed', False):
                    # engine-captured dense grad on a sparse-tagged param
                    # (sparse_engine._capture_hook dense branch): the
                    # engine applies the rows itself
                    continue
                if p.grad.layout != torch.strided:
                    # sparse grads are handled by the engine via
                    # sparse_row_step; a bare (non-engine) run coalesces here
                    if p.grad._nnz() == 0:
                        continue  # engine-swallowed (empty replacement grad)
                    g = p.grad.coalesce()
                    self.sparse_row_step(p, g.indices().flatten(),
                                         g.values(), group)
                    continue
                params.append(p)
                grads.append(p.grad)
            if params:
                self._dense_step(params, grads, group)
        return loss

    def _dense_step(self, params, grads, group):
        raise NotImplementedError

    def sparse_row_step(self, param, ids, grad_rows, group=None):
        raise NotImplementedError

    # -- fp32 master weights (pure-bf16 training) ----------------------------
    def _ensure_masters(self, params):
        """fp32 master copies for non-fp32 params: the fused kernels read/
        update the master and write the rounded copy into the bf16 param,
        so a pure-bf16 model trains with fp32 update precision and no
        per-step cast kernels.  Returns (masters, has_any): an empty list
        when every param is fp32 (zero-overhead common case), else one
        entry per param (the param itself stands in for fp32 entries)."""
        if not any(p.dtype != torch.float32 for p in params):
            return [], False
        out = []
        for p in params:
            if p.dtype == torch.float32:
                out.append(p)
                continue
            st = self.state[p]
            if 'master' not in st:
                m = torch.empty_like(p.detach(), dtype=torch.float32)
                m.copy_(p.detach())
                st['master'] = m
            out.append(st['master'])
        return out, True

    def load_state_dict(self, state_dict):
        """torch's default load casts floating state to the param's dtype —
        that would round fp32 masters/accumulators to bf16.  Restore state
        tensors verbatim (device move only)."""
        super().load_state_dict(state_dict)
        # undo any dtype casts torch applied: our state is authored at the
        # correct dtype, so restore from the raw payload
        id_map = {}
        for sg, g in zip(state_dict['param_groups'], self.param_groups):
            for old_id, p in zip(sg['params'], g['params']):
                id_map[old_id] = p
        for old_id, saved in state_dict['state'].items():
            p = id_map.get(old_id)
            if p is None:
                continue
            st = self.state[p]
            for k, v in saved.items():
                if torch.is_tensor(v) and k in st \
                        and torch.is_tensor(st[k]) \
                        and st[k].dtype != v.dtype:
                    st[k] = v.to(device=p.device)

    def group_of(self, param):
        for group in self.param_groups:
            for p in group['params']:
                if p is param:
                    return group
        raise KeyError('param not in optimizer')


class SGD(ParallaxOptimizer):
    """ApplyGradientDescent / ApplyMomentum (+nesterov)
    (reference graph_transform_lib.py:57,66)."""

    def __init__(self, params, lr, momentum=0.0, nesterov=False,
                 weight_decay=0.0):
        defaults = dict(lr=lr, momentum=momentum, nesterov=nesterov,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

    def _dense_step(self, params, grads, group):
        lr, mom = group['lr'], group['momentum']
        wd, nesterov = group['weight_decay'], group['nesterov']
        masters, has_m = self._ensure_masters(params)
        if mom == 0.0:
            if _use_native(params):
                for ps, gs, ms_ in _dtype_groups(params, list(grads),
                                                 masters or params):
                    _ops.ext().multi_tensor_sgd(
                        ps, gs, lr, wd, ms_ if masters else [])
                return
            for i, (p, g) in enumerate(zip(params, grads)):
                w = masters[i] if has_m else p
                g = g.to(w.dtype)
                if wd:
                    g = g.add(w, alpha=wd)
                w.add_(g, alpha=-lr)
                if w is not p:
                    p.copy_(w)
            return
        bufs = []
        for p in params:
            st = self.state[p]
            if 'momentum_buffer' not in st:
                st['momentum_buffer'] = torch.zeros_like(
                    p, dtype=torch.float32)
            bufs.append(st['momentum_buffer'])
        if _use_native(params):
            for ps, gs, bs, ms_ in _dtype_groups(params, list(grads),
                                                 bufs, masters or params):
                _ops.ext().multi_tensor_momentum(
                    ps, gs, bs, lr, mom, wd, nesterov,
                    ms_ if masters else [])
            return
        for i, (p, g, buf) in enumerate(zip(params, grads, bufs)):
            w = masters[i] if has_m else p
            g = g.to(buf.dtype)
            if wd:
                g = g.add(w.to(buf.dtype), alpha=wd)
            buf.mul_(mom).add_(g)
            upd = g.add(buf, alpha=mom) if nesterov else buf
            w.add_(upd.to(w.dtype), alpha=-lr)
            if w is not p:
                p.copy_(w)

    def supports_padded_sparse(self):
        return self.defaults.get('momentum', 0.0) == 0.0

    @torch.no_grad()
    def sparse_row_step(self, param, ids, grad_rows, group=None,
                        nvalid=None):
        group = group or self.group_of(param)
        lr = group['lr']
        if _use_native([param]):
            _ops.ext().sparse_sgd(param, ids.to(torch.int64),
                                  grad_rows.contiguous(), lr,
                                  nvalid=nvalid)
            return
        # index_add is a pure sum: padded zero-rows are no-ops, no mask
        param.index_add_(0, ids, grad_rows.to(param.dtype), alpha=-lr)


class Adagrad(ParallaxOptimizer):
    """ApplyAdagrad (dense) + SparseApplyAdagrad (sparse rows)
    (reference graph_transform_lib.py:59,74) — the LM1B optimizer
    (language_model_graph.py:60-61)."""

    def __init__(self, params, lr=0.2, initial_accumulator_value=1.0,
                 eps=1e-10):
        defaults = dict(lr=lr,
                        initial_accumulator_value=initial_accumulator_value,
                        eps=eps)
        super().__init__(params, defaults)

    def _accum(self, p, group):
        st = self.state[p]
        if 'sum' not in st:
            st['sum'] = torch.full_like(
                p, group['initial_accumulator_value'],
                dtype=torch.float32)
        return st['sum']

    def _dense_step(self, params, grads, group):
        lr, eps = group['lr'], group['eps']
        accums = [self._accum(p, group) for p in params]
        masters, has_m = self._ensure_masters(params)
        if _use_native(params):
            for ps, gs, acs, ms_ in _dtype_groups(params, list(grads),
                                                  accums,
                                                  masters or params):
                _ops.ext().multi_tensor_adagrad(
                    ps, gs, acs, lr, eps, ms_ if masters else [])
            return
        for i, (p, g, acc) in enumerate(zip(params, grads, accums)):
            w = masters[i] if has_m else p
            g32 = g.float()
            acc.add_(g32 * g32)
            w.add_((g32 / (acc.sqrt() + eps)).to(w.dtype), alpha=-lr)
            if w is not p:
                p.copy_(w)

    def supports_padded_sparse(self):
        return True

    @torch.no_grad()
    def sparse_row_step(self, param, ids, grad_rows, group=None,
                        nvalid=None):
        group = group or self.group_of(param)
        lr, eps = group['lr'], group['eps']
        acc = self._accum(param, group)
        if _use_native([param]):
            _ops.ext().sparse_adagrad(param, acc, ids.to(torch.int64),
                                      grad_rows.contiguous(), lr, eps,
                                      nvalid=nvalid)
            return
        # sum-style index ops: padded zero-rows add zero everywhere (the
        # duplicate-id-0 tail is benign without masking on this path)
        g32 = grad_rows.float()
        acc.index_add_(0, ids, g32 * g32)
        denom = acc.index_select(0, ids).sqrt() + eps
        param.index_add_(0, ids, (g32 / denom).to(param.dtype), alpha=-lr)


class Adam(ParallaxOptimizer):
    """ApplyAdam (reference graph_transform_lib.py:67) — skip_thoughts
    optimizer."""

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        # hipGraph-capturable dense path: one 0-dim fp32 device step
        # counter per param group, shared by its dense params (the dense
        # engine guarantees they all step together); the kernel
        # dereferences it at run time, so a captured optimizer graph
        # replays with fresh bias correction.
        self._dev_steps = {}
        self._dev_step_params = set()

    def _state(self, p):
        st = self.state[p]
        if 'step' not in st:
            st['step'] = 0
            st['exp_avg'] = torch.zeros_like(p, dtype=torch.float32)
            st['exp_avg_sq'] = torch.zeros_like(p, dtype=torch.float32)
        return st

    # -- device-step bookkeeping (checkpoint keeps integer 'step') ----------
    def _sync_steps_from_device(self):
        """Under graph replay only the device counter advances; fold it
        back into the per-param integer 'step' before serializing."""
        for gi, buf in self._dev_steps.items():
            t = int(buf.item())
            for p in self.param_groups[gi]['params']:
                if p in self._dev_step_params and p in self.state:
                    self.state[p]['step'] = t

    def state_dict(self):
        self._sync_steps_from_device()
        return super().state_dict()

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        self._dev_steps = {}  # re-seeded from restored steps on next step
        self._dev_step_params = set()

    def _dense_step(self, params, grads, group):
        lr, (b1, b2), eps = group['lr'], group['betas'], group['eps']
        wd = group['weight_decay']
        ms, vs, steps = [], [], []
        for p in params:
            st = self._state(p)
            st['step'] += 1
            ms.append(st['exp_avg'])
            vs.append(st['exp_avg_sq'])
            steps.append(st['step'])
        masters, has_m = self._ensure_masters(params)
        if _use_native(params):
            gi = next(i for i, g in enumerate(self.param_groups)
                      if g is group)
            buf = self._dev_steps.get(gi)
            if buf is None:
                buf = torch.tensor(float(steps[0] - 1), dtype=torch.float32,
                                   device=params[0].device)
                self._dev_steps[gi] = buf
            self._dev_step_params.update(params)
            buf.add_(1.0)
            for ps, gs, m_, v_, ms_ in _dtype_groups(
                    params, list(grads), ms, vs, masters or params):
                _ops.ext().multi_tensor_adam_dev(
                    ps, gs, m_, v_, buf, lr, b1, b2, eps, wd,
                    ms_ if masters else [])
            return
        for i, (p, g, m, v, t) in enumerate(zip(params, grads, ms, vs,
                                                steps)):
            w = masters[i] if has_m else p
            g32 = g.float()
            if wd:
                g32 = g32.add(w.float(), alpha=wd)
            m.mul_(b1).add_(g32, alpha=1 - b1)
            v.mul_(b2).addcmul_(g32, g32, value=1 - b2)
            bc1 = 1 - b1 ** t
            bc2 = 1 - b2 ** t
            step_size = lr / bc1
            denom = (v / bc2).sqrt().add_(eps)
            w.add_((m / denom).to(w.dtype), alpha=-step_size)
            if w is not p:
                p.copy_(w)

    @torch.no_grad()
    def sparse_row_step(self, param, ids, grad_rows, group=None,
                        nvalid=None):
        """Lazy-Adam style: moments updated only on touched rows (the TF
        sparse Adam apply)."""
        assert nvalid is None, 'Adam opts out of padded sparse inputs'
        group = group or self.group_of(param)
        lr, (b1, b2), eps = group['lr'], group['betas'], group['eps']
        st = self._state(param)
        st['step'] += 1
        t = st['step']
        m, v = st['exp_avg'], st['exp_avg_sq']
        g32 = grad_rows.float()
        m_rows = m.index_select(0, ids).mul_(b1).add_(g32, alpha=1 - b1)
        v_rows = v.index_select(0, ids).mul_(b2).addcmul_(g32, g32,
                                                          value=1 - b2)
        m.index_copy_(0, ids, m_rows)
        v.index_copy_(0, ids, v_rows)
        bc1, bc2 = 1 - b1 ** t, 1 - b2 ** t
        denom = (v_rows / bc2).sqrt().add_(eps)
        upd = (m_rows / bc1 / denom).to(param.dtype)
        param.index_add_(0, ids, upd, alpha=-lr)


class RMSProp(ParallaxOptimizer):
    """ApplyRMSProp (reference graph_transform_lib.py:68) — tf_cnn_benchmarks
    option (benchmark_cnn.py:805-818)."""

    def __init__(self, params, lr, decay=0.9, momentum=0.9, eps=1.0):
        defaults = dict(lr=lr, decay=decay, momentum=momentum, eps=eps)
        super().__init__(params, defaults)

    def _dense_step(self, params, grads, group):
        lr, decay = group['lr'], group['decay']
        mom, eps = group['momentum'], group['eps']
        rms, bufs = [], []
        for p in params:
            st = self.state[p]
            if 'square_avg' not in st:
                st['square_avg'] = torch.zeros_like(p, dtype=torch.float32)
                st['momentum_buffer'] = torch.zeros_like(
                    p, dtype=torch.float32)
            rms.append(st['square_avg'])
            bufs.append(st['momentum_buffer'])
        masters, has_m = self._ensure_masters(params)
        if _use_native(params):
            for ps, gs, rs, bs, ms_ in _dtype_groups(
                    params, list(grads), rms, bufs, masters or params):
                _ops.ext().multi_tensor_rmsprop(
                    ps, gs, rs, bs, lr, decay, mom, eps,
                    ms_ if masters else [])
            return
        for i, (p, g, sq, buf) in enumerate(zip(params, grads, rms, bufs)):
            w = masters[i] if has_m else p
            g32 = g.float()
            sq.mul_(decay).addcmul_(g32, g32, value=1 - decay)
            upd = g32 / (sq.sqrt() + eps)
            buf.mul_(mom).add_(upd)
            w.add_(buf.to(w.dtype), alpha=-lr)
            if w is not p:
                p.copy_(w)

    @torch.no_grad()
    def sparse_row_step(self, param, ids, grad_rows, group=None,
                        nvalid=None):
        assert nvalid is None, 'RMSProp opts out of padded sparse inputs'
        group = group or self.group_of(param)
        lr, decay = group['lr'], group['decay']
        mom, eps = group['momentum'], group['eps']
        st = self.state[param]
        if 'square_avg' not in st:
            st['square_avg'] = torch.zeros_like(param, dtype=torch.float32)
        sq = st['square_avg']
        g32 = grad_rows.float()
        sq_rows = sq.index_select(0, ids).mul_(decay).addcmul_(
            g32, g32, value=1 - decay)
        sq.index_copy_(0, ids, sq_rows)
        upd = (g32 / (sq_rows.sqrt() + eps)).to(param.dtype)
        param.index_add_(0, ids, upd, alpha=-lr)
