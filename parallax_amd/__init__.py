"""parallax_amd — an MI355X-native sparsity-aware data-parallel training
framework.

A from-scratch re-design of the capabilities of snuspl/parallax for a
single 8xMI355X node: PyTorch-ROCm + hand-written HIP/CDNA4 kernels + RCCL
over xGMI.  The public surface mirrors the reference
(parallax/parallax/__init__.py:16-26): ``parallel_run``,
``get_partitioner``, ``shard``, ``log`` and the config-class tree.
"""

import os as _os

# Ship the exhaustively-tuned MIOpen user DB (miopen_udb/, produced by
# MIOPEN_FIND_ENFORCE=4 on an MI355X: ResNet-50 4.9k -> 5.36k img/s) with
# the tree, and point MIOpen at it before its first use.  An explicit
# MIOPEN_USER_DB_PATH in the environment wins.
_udb = _os.path.join(_os.path.dirname(_os.path.dirname(
    _os.path.abspath(__file__))), 'miopen_udb')
if _os.path.isdir(_udb):
    _os.environ.setdefault('MIOPEN_USER_DB_PATH', _udb)

from parallax_amd.partitions import get_partitioner
from parallax_amd.runner import parallel_run
from parallax_amd import shard
from parallax_amd.lib import parallax_log as log

from parallax_amd.config import ParallaxConfig as Config
from parallax_amd.config import PSConfig
from parallax_amd.config import MPIConfig
from parallax_amd.config import CommunicationConfig
from parallax_amd.config import CheckPointConfig
from parallax_amd.config import ProfileConfig

from parallax_amd.session import TrainGraph
from parallax_amd.tagging import mark_sparse, mark_dense

__version__ = '0.1.0'

__all__ = [
    'parallel_run', 'get_partitioner', 'shard', 'log',
    'Config', 'ParallaxConfig', 'PSConfig', 'MPIConfig',
    'CommunicationConfig', 'CheckPointConfig', 'ProfileConfig',
    'TrainGraph', 'mark_sparse', 'mark_dense',
]
