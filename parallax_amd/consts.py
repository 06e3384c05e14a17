"""Cross-process role-signalling protocol (env vars) and framework constants.

MI355X-native re-design of the reference env-var protocol
(reference: parallax/core/python/common/consts.py:18-38).  The names are kept
so the operational contract (master -> worker role inheritance across process
spawn) is recognizable, but the launch plane is a local per-GPU fork +
torch.distributed rendezvous instead of ssh/mpirun.
"""

import os

# --- Role signalling (reference consts.py:18-27) -------------------------
PARALLAX_RUN_OPTION = "PARALLAX_RUN_OPTION"
PARALLAX_RUN_MASTER = "PARALLAX_RUN_MASTER"
PARALLAX_RUN_MPI = "PARALLAX_RUN_MPI"
PARALLAX_RUN_PS = "PARALLAX_RUN_PS"
PARALLAX_RUN_HYBRID = "PARALLAX_RUN_HYBRID"

PARALLAX_WORKER_ID = "PARALLAX_WORKER_ID"
PARALLAX_NUM_WORKERS = "PARALLAX_NUM_WORKERS"
PARALLAX_RESOURCE_INFO = "PARALLAX_RESOURCE_INFO"
PARALLAX_MACHINE_ID = "PARALLAX_MACHINE_ID"
PARALLAX_HOSTNAME = "PARALLAX_HOSTNAME"
PARALLAX_SYNC = "PARALLAX_SYNC"

# --- Partition search (reference partitions.py:29-31) ---------------------
PARALLAX_MIN_PARTITIONS = "PARALLAX_MIN_PARTITIONS"
PARALLAX_PARTITIONS = "PARALLAX_PARTITIONS"
PARALLAX_SEARCH = "PARALLAX_SEARCH"

# --- torch.distributed rendezvous -----------------------------------------
MASTER_ADDR = "MASTER_ADDR"
MASTER_PORT = "MASTER_PORT"
DEFAULT_MASTER_ADDR = "127.0.0.1"

# --- Local staging dir (reference consts.py:33-35) -------------------------
REMOTE_PARALLAX_ROOT = os.path.join(
    "/tmp", "parallax-%s" % os.environ.get("USER", "root"))

# --- Measurement windows (reference consts.py:37-38,
#     session_context.py:28-29) ---------------------------------------------
NUM_ITERATIONS_FOR_TEST = 200
NUM_ITERATIONS_FOR_WARMUP = 200
# step window timed for partition search (reference
# session_context.py:28-29); env-overridable for fast harness tests
COLLECT_STAT_START_STEP = int(os.environ.get('PARALLAX_COLLECT_STAT_START',
                                             '50'))
COLLECT_STAT_END_STEP = int(os.environ.get('PARALLAX_COLLECT_STAT_END',
                                           '100'))
