"""User-facing configuration tree.

Mirrors the reference's config-class surface
(parallax/core/python/common/config.py:21-179) so a parallax user finds the
same knobs, re-interpreted for the MI355X single-node architecture:

* PSConfig        -- in-process parameter-server (sharded sparse tables in
                     each rank's HBM) options; ``protocol`` is kept for API
                     parity but the transport is always RCCL over xGMI.
* MPIConfig       -- launch-plane extras; mpirun does not exist here, the
                     field configures the local per-GPU spawner.
* CheckPointConfig / ProfileConfig / CommunicationConfig / ParallaxConfig --
                     same fields and defaults as the reference.
"""


class PSConfig(object):
    """Options for the sparse (parameter-server) plane.

    Reference: config.py:21-49.  On MI355X the "servers" are the training
    ranks themselves: each embedding shard lives in one rank's HBM3E and is
    updated in place; ``protocol`` is accepted for compatibility and must be
    'rccl' or 'grpc' (both map to the in-process RCCL/xGMI exchange).
    """

    def __init__(self,
                 protocol='rccl',
                 replicate_variables=True,
                 local_aggregation=True,
                 boundary_among_servers=True,
                 boundary_between_workers_and_servers=True):
        if protocol not in ('rccl', 'grpc', 'grpc+verbs', 'grpc+gdr'):
            # reference PSConfig.protocol vocabulary (config.py:21-49);
            # every value maps to the in-process RCCL/xGMI exchange here
            raise ValueError('unknown PS protocol %r' % (protocol,))
        self.protocol = protocol
        self.replicate_variables = replicate_variables
        self.local_aggregation = local_aggregation
        self.boundary_among_servers = boundary_among_servers
        self.boundary_between_workers_and_servers = \
            boundary_between_workers_and_servers


class MPIConfig(object):
    """Options for the dense (all-reduce) plane launcher.

    Reference: config.py:51-69.  ``mpirun_options`` is parsed for parity but
    ignored by the local spawner (there is no mpirun on a single node).
    """

    def __init__(self, mpirun_options=''):
        self.mpirun_options = self.parse_mpirun_options(mpirun_options)

    def parse_mpirun_options(self, mpirun_options):
        if isinstance(mpirun_options, str):
            return mpirun_options
        elif isinstance(mpirun_options, list):
            return ' '.join([str(option) for option in mpirun_options])
        else:
            raise ValueError(
                'mpirun_options should be a string or a list of strings')


class CommunicationConfig(object):
    """Reference: config.py:72-81."""

    def __init__(self, ps_config=None, mpi_config=None,
                 dense_bucket_bytes=50 * 1024 * 1024,
                 overlap_grad_comm=True):
        self.ps_config = ps_config if ps_config is not None else PSConfig()
        self.mpi_config = mpi_config if mpi_config is not None else MPIConfig()
        assert isinstance(self.ps_config, PSConfig)
        assert isinstance(self.mpi_config, MPIConfig)
        # MI355X additions: all-reduce fusion-bucket size (sized for 7 xGMI
        # links x ~153 GB/s -> large buckets amortize per-link ring latency)
        # and backward/comm overlap toggle.
        self.dense_bucket_bytes = dense_bucket_bytes
        self.overlap_grad_comm = overlap_grad_comm
        # 'bf16' reduces dense grads in bf16 fusion buffers (halved xGMI
        # bytes; fp32 write-back); None keeps the gradient dtype.
        self.dense_comm_dtype = None


class CheckPointConfig(object):
    """Reference: config.py:84-99."""

    def __init__(self, ckpt_dir=None, save_ckpt_steps=None,
                 save_ckpt_secs=None):
        self.ckpt_dir = ckpt_dir
        self.save_ckpt_steps = save_ckpt_steps
        self.save_ckpt_secs = save_ckpt_secs


class ProfileConfig(object):
    """Reference: config.py:101-117.

    profile_dir:    directory for per-step traces (torch.profiler json +
                    kernel tables); the rocprofv3 analog of the reference's
                    RunMetadata dumps.
    profile_steps:  list of global steps to trace.
    profile_range:  (start, end) tuple of steps to trace.
    profile_worker: which rank traces (one tracer per machine, reference
                    mpi/runner.py:169-176).
    """

    def __init__(self, profile_dir=None, profile_steps=None,
                 profile_range=None, profile_worker=None):
        self.profile_dir = profile_dir
        self.profile_steps = profile_steps
        self.profile_range = profile_range
        self.profile_worker = profile_worker


class ParallaxConfig(object):
    """Configurable options of the framework.

    Reference: config.py:119-179 -- same fields, same defaults
    (run_option='HYBRID', average_sparse=False, search_partitions=True).
    ``sess_config`` is kept for parity; it may carry a dict of torch options
    (e.g. {'allow_tf32': False}).
    """

    def __init__(self,
                 run_option='HYBRID',
                 average_sparse=False,
                 sess_config=None,
                 redirect_path=None,
                 search_partitions=True,
                 export_graph_path=None,
                 communication_config=None,
                 ckpt_config=None,
                 profile_config=None):
        self.run_option = run_option
        self.average_sparse = average_sparse
        self.sess_config = sess_config
        self.redirect_path = redirect_path
        self.search_partitions = search_partitions
        self.export_graph_path = export_graph_path

        self.communication_config = (communication_config
                                     if communication_config is not None
                                     else CommunicationConfig())
        self.ckpt_config = (ckpt_config if ckpt_config is not None
                            else CheckPointConfig())
        self.profile_config = (profile_config if profile_config is not None
                               else ProfileConfig())

        self._sync = None
        self._resource_info = None

    def get_ckpt_config(self):
        return self.ckpt_config

    def set_sync(self, sync):
        self._sync = sync

    @property
    def sync(self):
        return self._sync

    def set_resource_info(self, resource_info):
        self._resource_info = resource_info

    @property
    def resource_info(self):
        return self._resource_info
