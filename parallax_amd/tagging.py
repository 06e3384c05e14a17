"""Dense/sparse gradient classification — the GradientsInfo equivalent.

The reference's TF fork records, for every trainable variable, whether its
gradient is a dense Tensor or an IndexedSlices (embedding rows) in a custom
``GRADIENTS_INFO`` graph collection (contract N1; reference
common/runner.py:40-60, graph_transform_lib.py:43,453-480).  That record is
what drives the whole MPI/PS/HYBRID dispatch.

MI355X-native equivalent: an autograd-level tagging pass over a PyTorch
module.  A parameter is SPARSE when its gradient arrives as indexed slices —
which in our framework means it is the weight of an embedding-style lookup
(nn.Embedding / nn.EmbeddingBag / our ParallaxEmbedding, or a parameter the
user marked explicitly).  Everything else is DENSE.

Two classification mechanisms, strongest wins:
 1. explicit  — ``mark_sparse(param)`` / ``mark_dense(param)``;
 2. structural — module scan for embedding-type modules (+ partitioned
    ParallaxEmbedding shard lists).
A third, dynamic probe (run one fwd/bwd and inspect ``grad.layout``) is
available as ``classify_by_probe`` for models whose structure hides the
lookup.
"""

from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.nn as nn

_PARALLAX_TAG_ATTR = '_parallax_grad_tag'

DENSE = 'dense'
SPARSE = 'sparse'


def mark_sparse(param):
    """Explicitly tag a parameter as receiving indexed-slices gradients."""
    setattr(param, _PARALLAX_TAG_ATTR, SPARSE)
    return param


def mark_dense(param):
    setattr(param, _PARALLAX_TAG_ATTR, DENSE)
    return param


@dataclass
class GradientsInfo:
    """Per-parameter record (reference GradientsInfoDef protobuf, N1).

    name:       user-visible parameter name (checkpoint key).
    param:      the tensor.
    is_sparse:  True == INDEXED_SLICES (reference
                gradients_info_pb2.TensorInfoDef.INDEXED_SLICES).
    num_rows:   embedding row count (sparse only).
    partition:  (num_partitions, row_offsets) when the parameter is a
                virtually partitioned table (reference
                tf.fixed_size_partitioner shards 'emb/part_k').
    """
    name: str
    param: torch.nn.Parameter
    is_sparse: bool
    num_rows: int = 0
    partition: Optional[tuple] = None


_EMBEDDING_TYPES = (nn.Embedding, nn.EmbeddingBag)


def _is_embedding_module(module):
    if isinstance(module, _EMBEDDING_TYPES):
        return True
    # our own ops (ParallaxEmbedding / PartitionedEmbedding) self-describe
    return getattr(module, '_parallax_is_embedding', False)


def classify_params(model: nn.Module) -> List[GradientsInfo]:
    """Structural + explicit classification pass.

    Returns one GradientsInfo per trainable parameter, in deterministic
    ``named_parameters`` order (that order also fixes the collective
    bucketing order across ranks — contract N6).
    """
    infos = []
    sparse_params = {}  # id(param) -> (num_rows, partition)
    for mod_name, module in model.named_modules():
        if _is_embedding_module(module):
            for p_name, p in module.named_parameters(recurse=True):
                if p.dim() >= 2:
                    part = getattr(p, '_parallax_partition_meta', None)
                    sparse_params[id(p)] = (p.shape[0], part)
    for name, p in model.named_parameters():
        if not p.requires_grad:
            continue
        explicit = getattr(p, _PARALLAX_TAG_ATTR, None)
        if explicit == SPARSE:
            infos.append(GradientsInfo(
                name, p, True, p.shape[0],
                getattr(p, '_parallax_partition_meta', None)))
        elif explicit == DENSE:
            infos.append(GradientsInfo(name, p, False))
        elif id(p) in sparse_params:
            num_rows, part = sparse_params[id(p)]
            infos.append(GradientsInfo(name, p, True, num_rows, part))
        else:
            infos.append(GradientsInfo(name, p, False))
    return infos


def classify_by_probe(model: nn.Module, probe_fn) -> List[GradientsInfo]:
    """Dynamic classification: run ``probe_fn()`` (one forward+backward on
    dummy data) and tag parameters whose ``.grad`` is a sparse COO tensor.
    Reference analog: the TF fork's tf.gradients recording IndexedSlices
    (N1).  Clears grads afterwards."""
    model.zero_grad(set_to_none=True)
    probe_fn()
    infos = []
    for name, p in model.named_parameters():
        if not p.requires_grad:
            continue
        is_sparse = p.grad is not None and p.grad.layout != torch.strided
        infos.append(GradientsInfo(name, p, is_sparse,
                                   p.shape[0] if is_sparse else 0))
    model.zero_grad(set_to_none=True)
    return infos


def split_dense_sparse(infos):
    """Reference runner.py:40-60 ``_get_grads`` split."""
    sparse = [gi for gi in infos if gi.is_sparse]
    dense = [gi for gi in infos if not gi.is_sparse]
    assert len(sparse) > 0 or len(dense) > 0
    return sparse, dense
