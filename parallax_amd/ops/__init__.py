"""parallax_amd.ops — hand-written CDNA4 HIP kernels + CPU reference impls.

Dispatch policy (per the framework's no-silent-fallback rule):
  * CUDA/HIP tensors  -> the native extension ``parallax_amd._C`` MUST be
    importable; if it is missing the op RAISES instead of silently falling
    back to eager PyTorch.  (Set PARALLAX_ALLOW_EAGER_GPU=1 to override for
    debugging only.)
  * CPU tensors       -> plain PyTorch reference implementations (also the
    ground truth for the numerics tests in tests/test_ops_*.py).

The kernels themselves live in parallax_amd/ops/hip/ and are built in-tree
for gfx950 via ``python -m parallax_amd.ops.build`` (or setup.py / the
repo-level __graft_entry__.build()).
"""

import os

import torch

_EXT = None
_EXT_ERR = None


def _try_load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    import glob
    import importlib.util
    lib_dir = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), '_lib')
    sos = glob.glob(os.path.join(lib_dir, 'parallax_amd_C*.so'))
    if not sos:
        _EXT_ERR = ImportError('no parallax_amd_C*.so under %s' % lib_dir)
        return None
    try:
        spec = importlib.util.spec_from_file_location('parallax_amd_C',
                                                      sos[0])
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _EXT = mod
    except Exception as e:  # pragma: no cover - load failure on GPU box
        _EXT_ERR = e
        _EXT = None
    return _EXT


def native_available():
    return _try_load_ext() is not None


def ext():
    """Return the native extension module, raising loudly on a GPU box where
    it is missing (a GPU test passing on an eager fallback is a bug)."""
    m = _try_load_ext()
    if m is None:
        if os.environ.get('PARALLAX_ALLOW_EAGER_GPU') == '1':
            return None
        raise RuntimeError(
            'parallax_amd native extension (parallax_amd._C) is not built; '
            'run `python -m parallax_amd.ops.build` (gfx950). '
            'Original import error: %r' % (_EXT_ERR,))
    return m


def use_native(*tensors):
    """True when the op should dispatch to the HIP kernel."""
    on_gpu = any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if not on_gpu:
        return False
    ext()  # raises if missing (unless eager override)
    return _EXT is not None


from parallax_amd.ops.sparse import (unique_sum_rows, segment_sum_rows,
                                     rows_gather)
from parallax_amd.ops.embedding import (ParallaxEmbedding,
                                        PartitionedEmbedding,
                                        embedding_lookup)
from parallax_amd.ops.layernorm import layer_norm, ParallaxLayerNorm
from parallax_amd.ops.softmax_ce import softmax_cross_entropy
from parallax_amd.ops.lstm import lstm_cell, ParallaxLSTM
from parallax_amd.ops.sampled_softmax import sampled_softmax_loss
from parallax_amd.ops.misc import (clip_by_global_norm, global_norm,
                                   ema_update)
from parallax_amd.ops.linear import PxLinear
from parallax_amd.ops.attention import px_attention
from parallax_amd.ops.batchnorm import ParallaxBatchNorm2d
from parallax_amd.ops.pool import ParallaxMaxPool3x3s2, ParallaxGlobalAvgPool
from parallax_amd.ops.conv import (ParallaxConv1x1, ParallaxConv3x3,
                                   swap_conv1x1, swap_conv3x3)
