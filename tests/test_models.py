"""Model-level tests: BASELINE config 1 (ResNet-50 hybrid world_size=1 on
CPU/gloo, synthetic 224x224) and LM1B/model construction checks."""

import os

import pytest
import torch

import parallax_amd
from parallax_amd.config import ParallaxConfig
from parallax_amd.models.resnet import (ResNet, build_resnet_graph,
                                        synthetic_batch)
from parallax_amd.models.lm1b import LM1BModel, build_lm1b_graph
from parallax_amd.models import lm1b as lm1b_mod


@pytest.fixture(autouse=True)
def _destroy_pg():
    yield
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()


def _worker_env(monkeypatch, port):
    monkeypatch.setenv('WORLD_SIZE', '1')
    monkeypatch.setenv('RANK', '0')
    monkeypatch.setenv('MASTER_ADDR', '127.0.0.1')
    monkeypatch.setenv('MASTER_PORT', str(port))


def test_resnet50_construction():
    m = ResNet(50, num_classes=1001)
    n_params = sum(p.numel() for p in m.parameters())
    # ResNet-50 w/ 1001 classes: ~25.6M params (reference ~25.5M + fc)
    assert 24e6 < n_params < 27e6
    x = torch.randn(2, 3, 64, 64)  # small spatial for CPU speed
    out = m(x)
    assert out.shape == (2, 1001)


def test_baseline_config1_resnet_hybrid_cpu(monkeypatch, free_port):
    """BASELINE config 1: ResNet-50 hybrid-mode world_size=1 on CPU/gloo,
    synthetic 224x224 — one full engine step, finite loss, falls back to
    MPI (all-dense)."""
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)
    sess, num_workers, worker_id, n_rep = parallax_amd.parallel_run(
        lambda: build_resnet_graph(depth=50, batch_size=2, use_bf16=False,
                                   channels_last=False),
        None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    assert sess.engine.mode == 'MPI'  # all-dense degenerate fallback
    x, y = synthetic_batch(batch_size=2, image_size=224)
    out = sess.run((x, y))
    assert torch.isfinite(out['loss'])
    assert out['global_step'] == 1
    sess.close()


def test_lm1b_model_parameter_shapes():
    m = LM1BModel(vocab_size=1000, emb_dim=16, state_size=32, proj_size=16,
                  num_sampled=64, min_partitions=4)
    from parallax_amd.tagging import classify_params, split_dense_sparse
    sparse, dense = split_dense_sparse(classify_params(m))
    # sparse: emb table + softmax_w table + softmax_b
    assert len(sparse) == 3
    # dense: lstm w, b, w_proj
    assert len(dense) == 3
    names = {gi.name for gi in sparse}
    assert names == {'emb.weight', 'softmax_w.weight', 'softmax_b'}
    parts = {gi.name: gi.partition for gi in sparse}
    assert parts['emb.weight'][0] == 4  # min_partitions shards


def test_lm1b_partition_env_override(monkeypatch):
    monkeypatch.setenv('PARALLAX_PARTITIONS', '8')
    m = LM1BModel(vocab_size=1000, emb_dim=16, state_size=32, proj_size=16,
                  num_sampled=64, min_partitions=4)
    assert m.emb.num_partitions == 8


def test_lm1b_synthetic_batch_shapes():
    x, y, st = lm1b_mod.synthetic_batch(batch_size=4, num_steps=5,
                                        vocab_size=100)
    assert x.shape == (4, 5) and y.shape == (4, 5) and st is None
    assert int(x.max()) < 100


def test_lm1b_pure_bf16_learns(monkeypatch, free_port):
    """pure_bf16: bf16 LSTM body, fp32 tables/shadows, master-weight
    Adagrad — the eager CPU path must learn on a repeated batch."""
    monkeypatch.setenv('WORLD_SIZE', '1')
    monkeypatch.setenv('RANK', '0')
    monkeypatch.setenv('MASTER_ADDR', '127.0.0.1')
    monkeypatch.setenv('MASTER_PORT', str(free_port))
    torch.manual_seed(0)
    sess, *_ = parallax_amd.parallel_run(
        lambda: build_lm1b_graph(batch_size=4, num_steps=3, vocab_size=200,
                                 emb_dim=16, state_size=32, proj_size=16,
                                 num_sampled=32, min_partitions=4, lr=0.4,
                                 pure_bf16=True),
        None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    m = sess.graph.model
    assert m.lstm.w.dtype == torch.bfloat16
    assert m.emb.weight.dtype == torch.float32
    assert m.ema_0.dtype == torch.float32
    x = torch.randint(0, 200, (4, 3))
    y = torch.randint(0, 200, (4, 3))
    first = None
    for _ in range(30):
        out = sess.run((x, y, None))
        if first is None:
            first = float(out['loss'])
    assert float(out['loss']) < first
    assert sess.graph.optimizer.state[m.lstm.w]['master'].dtype \
        == torch.float32
    sess.close()
    import torch.distributed as dist
    dist.destroy_process_group()


def test_resnet_pure_bf16_structure_and_step(monkeypatch, free_port):
    """pure_bf16 ResNet: conv/FC weights bf16, BatchNorm fp32 (weights,
    stats); a training step runs and learns through the master-weight
    momentum path."""
    monkeypatch.setenv('WORLD_SIZE', '1')
    monkeypatch.setenv('RANK', '0')
    monkeypatch.setenv('MASTER_ADDR', '127.0.0.1')
    monkeypatch.setenv('MASTER_PORT', str(free_port))
    torch.manual_seed(0)
    import torch.nn as nn
    sess, *_ = parallax_amd.parallel_run(
        lambda: build_resnet_graph(depth=50, num_classes=10, lr=0.05,
                                   use_bf16=False, channels_last=False,
                                   pure_bf16=True),
        None, parallax_config=ParallaxConfig(run_option='MPI'))
    m = sess.graph.model
    convs = [mod for mod in m.modules() if isinstance(mod, nn.Conv2d)]
    from parallax_amd.ops.batchnorm import ParallaxBatchNorm2d
    bns = [mod for mod in m.modules()
           if isinstance(mod, (nn.modules.batchnorm._BatchNorm,
                               ParallaxBatchNorm2d))]
    assert convs and all(c.weight.dtype == torch.bfloat16 for c in convs)
    assert bns and all(b.weight.dtype == torch.float32 for b in bns)
    assert all(b.running_mean.dtype == torch.float32 for b in bns)
    x = torch.randn(2, 3, 64, 64)
    y = torch.randint(0, 10, (2,))
    losses = [float(sess.run((x, y))['loss']) for _ in range(5)]
    assert all(l == l for l in losses)
    assert losses[-1] < losses[0]
    st = sess.graph.optimizer.state[convs[0].weight]
    assert st['master'].dtype == torch.float32
    sess.close()
    import torch.distributed as dist
    dist.destroy_process_group()
