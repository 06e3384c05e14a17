"""PrefetchLoader (N9: StagingArea equivalent) tests — CPU passthrough
here; the async-copy path runs on GPU (covered by test_ops_gpu)."""

import torch

from parallax_amd.data import PrefetchLoader, sharded_prefetch_loader


def test_prefetch_passthrough_cpu():
    batches = [(torch.ones(2), torch.zeros(2)) for _ in range(4)]
    out = list(PrefetchLoader(batches, 'cpu'))
    assert len(out) == 4
    assert torch.equal(out[0][0], torch.ones(2))


def test_prefetch_dict_batches():
    batches = [{'x': torch.full((2,), float(i))} for i in range(3)]
    out = list(PrefetchLoader(batches, 'cpu'))
    assert [float(b['x'][0]) for b in out] == [0.0, 1.0, 2.0]


def test_sharded_prefetch():
    batches = [torch.tensor([i]) for i in range(10)]
    out = list(sharded_prefetch_loader(batches, 'cpu', num_shards=2,
                                       shard_id=1))
    assert [int(t) for t in out] == [1, 3, 5, 7, 9]
