"""Unit tests: config tree, resource_info codec, shard semantics,
partitioner math, tagging pass.  All CPU."""

import os

import pytest
import torch
import torch.nn as nn

import parallax_amd
from parallax_amd import lib, shard, tagging
from parallax_amd.config import (ParallaxConfig, PSConfig, MPIConfig,
                                 CommunicationConfig, CheckPointConfig,
                                 ProfileConfig)
from parallax_amd.partitions import (FixedSizePartitioner, fit_cost_model,
                                     PartitionStatCollector, get_partitioner)


# --- config (reference config.py defaults) ---------------------------------

def test_config_defaults():
    cfg = ParallaxConfig()
    assert cfg.run_option == 'HYBRID'
    assert cfg.average_sparse is False
    assert cfg.search_partitions is True
    assert isinstance(cfg.communication_config, CommunicationConfig)
    assert isinstance(cfg.communication_config.ps_config, PSConfig)
    assert cfg.communication_config.ps_config.local_aggregation is True
    assert isinstance(cfg.ckpt_config, CheckPointConfig)
    assert cfg.ckpt_config.ckpt_dir is None
    assert isinstance(cfg.profile_config, ProfileConfig)


def test_mpi_config_options_parse():
    assert MPIConfig(['-x', 'FOO=1']).mpirun_options == '-x FOO=1'
    assert MPIConfig('-np 4').mpirun_options == '-np 4'
    with pytest.raises(ValueError):
        MPIConfig(7)


# --- resource_info (reference lib.py:121-176) -------------------------------

def test_resource_info_parse_and_codec(tmp_path):
    p = tmp_path / 'resource_info'
    p.write_text('127.0.0.1:0,1,2,3\n')
    ri = lib.parse_resource_info(str(p), 'HYBRID')
    assert ri['worker'][0]['hostname'] == '127.0.0.1'
    assert ri['worker'][0]['gpus'] == [0, 1, 2, 3]
    # HYBRID reserves one port per GPU (reference lib.py:143-146)
    assert len(ri['worker'][0]['port']) == 4
    assert len(ri['master']) == 1 and len(ri['ps']) == 1

    s = lib.serialize_resource_info(ri)
    ri2 = lib.deserialize_resource_info(s)
    assert ri2 == ri


def test_resource_info_literal_string():
    ri = lib.parse_resource_info('localhost:0,1\n', 'MPI')
    assert ri['worker'][0]['gpus'] == [0, 1]
    assert len(ri['worker'][0]['port']) == 1
    assert lib.num_total_workers(ri) == 2


def test_resource_info_empty_raises():
    with pytest.raises(ValueError):
        lib.parse_resource_info('\n\n', 'MPI')


# --- shard (reference iterator_utils_test.py:109-173 semantics) -------------

def test_shard_filter_semantics():
    data = list(range(10))
    out = list(shard.shard(data, num_shards=2, shard_id=1))
    assert out == [1, 3, 5, 7, 9]
    out0 = list(shard.shard(data, num_shards=2, shard_id=0))
    assert out0 == [0, 2, 4, 6, 8]
    assert len(shard.shard(data, num_shards=3, shard_id=2)) == 3


def test_shard_env_protocol(monkeypatch):
    monkeypatch.setenv('PARALLAX_NUM_WORKERS', '4')
    monkeypatch.setenv('PARALLAX_WORKER_ID', '3')
    n, i = shard.create_num_shards_and_shard_id()
    assert (n, i) == (4, 3)
    out = list(shard.shard(list(range(8))))
    assert out == [3, 7]


def test_sharded_sampler():
    s = shard.ShardedSampler(10, num_shards=4, shard_id=1)
    assert list(s) == [1, 5, 9]
    assert len(s) == 3


# --- partitioner (reference partitions.py:35-51 + fixed_size_partitioner) ---

def test_fixed_size_partitioner_shards():
    p = FixedSizePartitioner(4)
    assert p.shard_sizes(10) == [3, 3, 2, 2]
    assert p.shard_offsets(10) == [0, 3, 6, 8]
    assert p(10) == [(0, 3), (3, 3), (6, 2), (8, 2)]
    # more partitions than rows: clamp
    assert FixedSizePartitioner(8).shard_sizes(3) == [1, 1, 1]


def test_get_partitioner_env(monkeypatch):
    monkeypatch.setenv('PARALLAX_PARTITIONS', '16')
    p = get_partitioner(4)
    assert p.num_partitions == 16
    monkeypatch.delenv('PARALLAX_PARTITIONS')
    monkeypatch.delenv('PARALLAX_MIN_PARTITIONS', raising=False)
    p = get_partitioner(4)
    assert p.num_partitions == 4
    assert os.environ['PARALLAX_MIN_PARTITIONS'] == '4'


def test_cost_model_fit_recovers_minimum():
    # t(n) = b/n + a(n-1) + c with known constants: argmin ~= sqrt(b/a)
    a, b, c = 0.5, 128.0, 3.0
    ps = [4, 8, 16, 32, 64]
    ts = [b / n + a * (n - 1) + c for n in ps]
    best = fit_cost_model(ps, ts)
    true_best = min(range(4, 65), key=lambda n: b / n + a * (n - 1) + c)
    assert abs(best - true_best) <= 1


def test_cost_model_fit_few_points():
    assert fit_cost_model([4, 8], [2.0, 1.0]) == 8


def test_stat_collector_walk():
    c = PartitionStatCollector(4, min_partitions=2)
    cont, p = c.update(10.0)          # first measurement -> double
    assert cont and p == 8
    cont, p = c.update(6.0)           # improved while increasing -> double
    assert cont and p == 16
    cont, p = c.update(9.0)           # worse while increasing -> go below min tested
    assert cont and p == 2
    cont, p = c.update(20.0)          # worse while decreasing -> stop + fit
    assert not cont
    assert p in range(2, 17)


def test_stat_collector_death_doubles():
    c = PartitionStatCollector(4, min_partitions=2)
    cont, p = c.update(None, alive=False)   # job died: p too small
    assert cont and p == 8
    assert c.min_partitions == 8


# --- tagging (contract N1) --------------------------------------------------

class _Toy(nn.Module):
    def __init__(self):
        super().__init__()
        self.emb = nn.Embedding(100, 8)
        self.fc = nn.Linear(8, 4)


def test_classify_params_structural():
    m = _Toy()
    infos = tagging.classify_params(m)
    by_name = {gi.name: gi for gi in infos}
    assert by_name['emb.weight'].is_sparse
    assert by_name['emb.weight'].num_rows == 100
    assert not by_name['fc.weight'].is_sparse
    assert not by_name['fc.bias'].is_sparse
    sparse, dense = tagging.split_dense_sparse(infos)
    assert len(sparse) == 1 and len(dense) == 2


def test_classify_params_explicit_marks():
    m = _Toy()
    tagging.mark_dense(m.emb.weight)
    tagging.mark_sparse(m.fc.weight)
    by_name = {gi.name: gi for gi in tagging.classify_params(m)}
    assert not by_name['emb.weight'].is_sparse
    assert by_name['fc.weight'].is_sparse


def test_classify_by_probe():
    from parallax_amd.ops.embedding import ParallaxEmbedding
    m = nn.Module()
    m.emb = ParallaxEmbedding(50, 4)
    m.lin = nn.Linear(4, 2)

    def probe():
        ids = torch.randint(0, 50, (6,))
        out = m.lin(m.emb(ids)).sum()
        out.backward()

    infos = tagging.classify_by_probe(m, probe)
    by_name = {gi.name: gi for gi in infos}
    assert by_name['emb.weight'].is_sparse
    assert not by_name['lin.weight'].is_sparse


def test_partitioned_embedding_tagging():
    from parallax_amd.ops.embedding import PartitionedEmbedding
    m = nn.Module()
    m.emb = PartitionedEmbedding(100, 8, FixedSizePartitioner(4))
    infos = tagging.classify_params(m)
    assert len(infos) == 1
    gi = infos[0]
    assert gi.is_sparse and gi.name == 'emb.weight'
    assert gi.num_rows == 100
    assert gi.partition == (4, (0, 25, 50, 75))


# --- public API surface (reference __init__.py:16-26) -----------------------

def test_public_api_surface():
    assert callable(parallax_amd.parallel_run)
    assert callable(parallax_amd.get_partitioner)
    assert hasattr(parallax_amd.shard, 'shard')
    assert parallax_amd.log is not None
    for name in ('Config', 'PSConfig', 'MPIConfig', 'CommunicationConfig',
                 'CheckPointConfig', 'ProfileConfig'):
        assert hasattr(parallax_amd, name)


def test_ps_protocol_validation():
    from parallax_amd.config import PSConfig
    for ok in ('rccl', 'grpc', 'grpc+verbs', 'grpc+gdr'):
        assert PSConfig(protocol=ok).protocol == ok
    with pytest.raises(ValueError):
        PSConfig(protocol='carrier-pigeon')


def test_stat_collector_persistent_deaths_raise():
    """Repeated deaths with no successful trial must not double the
    partition count forever (the failure is not partition-related)."""
    from parallax_amd.partitions import PartitionStatCollector
    c = PartitionStatCollector(4, min_partitions=1, max_trials=5)
    for _ in range(4):
        cont, p = c.update(None, alive=False)
        assert cont
    with pytest.raises(RuntimeError):
        c.update(None, alive=False)
