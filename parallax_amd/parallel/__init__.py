from parallax_amd.parallel.comm import CommContext
from parallax_amd.parallel.dense_engine import DenseAllReduceEngine
from parallax_amd.parallel.sparse_engine import (SparseAllGatherEngine,
                                                 SparsePSEngine)
from parallax_amd.parallel.hybrid import HybridEngine, resolve_run_option
