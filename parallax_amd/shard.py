"""Data sharding: every worker sees samples i where i % num_shards == shard_id.

MI355X-native equivalent of the reference's common/shard.py (87 loc).  The
reference appends ``_enumerate().filter(i % num_shards == shard_id)`` to a
tf.data pipeline with placeholder consts patched per worker
(shard.py:69-87); in eager PyTorch the same contract is a sampler/iterable
filter parameterized by env-derived (num_shards, shard_id), fixed at worker
start instead of graph-patch time.
"""

import os

from parallax_amd import consts


def create_num_shards_and_shard_id():
    """Return (num_shards, shard_id) for this worker process
    (reference shard.py:26-54: in-graph variables registered in collections;
    here: values resolved from the worker env protocol / torch.distributed).
    """
    num_shards = int(os.environ.get(consts.PARALLAX_NUM_WORKERS,
                                    os.environ.get('WORLD_SIZE', '1')))
    shard_id = int(os.environ.get(consts.PARALLAX_WORKER_ID,
                                  os.environ.get('RANK', '0')))
    if not 0 <= shard_id < max(num_shards, 1):
        raise ValueError('shard_id %d out of range for %d shards'
                         % (shard_id, num_shards))
    return num_shards, shard_id


class _ShardedIterable(object):
    """Iterable view keeping only elements with index % num_shards == shard_id
    (exact reference filter semantics, shard.py:69-87)."""

    def __init__(self, base, num_shards, shard_id):
        self.base = base
        self.num_shards = num_shards
        self.shard_id = shard_id

    def __iter__(self):
        for i, item in enumerate(self.base):
            if i % self.num_shards == self.shard_id:
                yield item

    def __len__(self):
        n = len(self.base)
        # number of i in [0, n) with i % num_shards == shard_id
        return (n - self.shard_id + self.num_shards - 1) // self.num_shards


class ShardedSampler(object):
    """torch Sampler yielding dataset indices i with
    i % num_shards == shard_id.
    Usable as DataLoader(sampler=ShardedSampler(len(ds)))."""

    def __init__(self, num_samples, num_shards=None, shard_id=None):
        if num_shards is None or shard_id is None:
            num_shards, shard_id = create_num_shards_and_shard_id()
        self.num_samples = num_samples
        self.num_shards = num_shards
        self.shard_id = shard_id

    def __iter__(self):
        return iter(range(self.shard_id, self.num_samples, self.num_shards))

    def __len__(self):
        n, s, k = self.num_samples, self.shard_id, self.num_shards
        return (n - s + k - 1) // k if n > s else 0


def shard(ds, num_shards=None, shard_id=None):
    """Shard a dataset/iterable across workers (reference shard.py:69-87).

    Accepts any iterable (including a torch IterableDataset or a generator
    factory) and returns an iterable seen by THIS worker only.  With
    explicit (num_shards, shard_id) the env protocol is bypassed — that is
    what the reference's patched consts do per worker.
    """
    if num_shards is None or shard_id is None:
        num_shards, shard_id = create_num_shards_and_shard_id()
    return _ShardedIterable(ds, num_shards, shard_id)
