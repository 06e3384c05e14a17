"""Gradient clipping, global norm, EMA — the remaining §2.4 glue kernels.

Reference analogs: clip_by_global_norm over ~20M LSTM params
(examples/lm1b/language_model_graph.py:53), clip_by_value
(benchmark_cnn.py:797-800), ExponentialMovingAverage decay .999
(language_model_graph.py:63-65).

GPU path: multi-tensor fused L2-norm + scale kernels (ops/hip/multi_tensor.hip)
— one launch over all grads instead of per-tensor reduce chains.
"""

import torch

from parallax_amd import ops as _ops


def _dtype_groups(tensors):
    """Group tensors by dtype, preserving order.  The multi-tensor kernels
    dispatch on tensors[0].scalar_type() and assume a uniform list — a
    mixed bf16+fp32 list would reinterpret memory."""
    groups = {}
    for t in tensors:
        groups.setdefault(t.dtype, []).append(t)
    return list(groups.values())


def global_norm(tensors):
    """sqrt(sum ||t||^2) over a tensor list (fp32 accumulation)."""
    tensors = [t for t in tensors if t is not None]
    if not tensors:
        return torch.tensor(0.0)
    if _ops.use_native(*tensors):
        sq = None
        for grp in _dtype_groups(tensors):
            n = _ops.ext().multi_tensor_l2norm(grp)
            sq = n * n if sq is None else sq + n * n
        return sq.sqrt()
    acc = torch.zeros((), dtype=torch.float32, device=tensors[0].device)
    for t in tensors:
        acc = acc + t.detach().float().pow(2).sum()
    return acc.sqrt()


def clip_by_global_norm(tensors, clip_norm):
    """In-place scale of every tensor by clip_norm/max(norm, clip_norm);
    returns the pre-clip global norm (TF clip_by_global_norm semantics)."""
    tensors = [t for t in tensors if t is not None]
    norm = global_norm(tensors)
    if _ops.use_native(*tensors):
        for grp in _dtype_groups(tensors):
            _ops.ext().multi_tensor_scale_by_norm(grp, norm,
                                                  float(clip_norm))
        return norm
    scale = clip_norm / torch.clamp(norm, min=clip_norm)
    for t in tensors:
        t.detach().mul_(scale.to(t.dtype))
    return norm


def ema_update(shadow_tensors, tensors, decay):
    """shadow = decay * shadow + (1-decay) * value (TF EMA apply).
    Shadows may be fp32 over bf16 values (pure-bf16 bodies keep fp32
    shadows so the .999-decay drift is not swallowed by bf16 rounding)."""
    shadow_tensors = list(shadow_tensors)
    tensors = [t.to(s.dtype) if t.dtype != s.dtype else t
               for s, t in zip(shadow_tensors, tensors)]
    if _ops.use_native(*tensors):
        _ops.ext().multi_tensor_ema(shadow_tensors, tensors, float(decay))
        return
    with torch.no_grad():
        for s, t in zip(shadow_tensors, tensors):
            s.mul_(decay).add_(t, alpha=1.0 - decay)
