"""World-size-1 engine/session tests on CPU (gloo) — the full worker path
of parallel_run without a spawner."""

import os

import pytest
import torch
import torch.nn as nn

import parallax_amd
from parallax_amd.config import ParallaxConfig, CheckPointConfig
from parallax_amd.models.simple import build_simple_graph
from parallax_amd.models.lm1b import build_lm1b_graph, synthetic_batch
from parallax_amd.runner import _run_worker
from parallax_amd.session import TrainGraph
from parallax_amd import optim
from parallax_amd.ops.embedding import ParallaxEmbedding


def _worker_env(monkeypatch, free_port, world=1, rank=0):
    monkeypatch.setenv('WORLD_SIZE', str(world))
    monkeypatch.setenv('RANK', str(rank))
    monkeypatch.setenv('MASTER_ADDR', '127.0.0.1')
    monkeypatch.setenv('MASTER_PORT', str(free_port))


@pytest.fixture(autouse=True)
def _destroy_pg():
    yield
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()


def test_simple_loss_decreases(monkeypatch, free_port):
    """The reference 'simple' example contract: loss decreases
    (simple_driver.py)."""
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)
    sess, num_workers, worker_id, n_rep = parallax_amd.parallel_run(
        lambda: build_simple_graph(lr=0.05), resource_info=None,
        parallax_config=ParallaxConfig(run_option='HYBRID'))
    assert (num_workers, worker_id, n_rep) == (1, 0, 1)
    x = torch.randn(16, 2)
    y = x @ torch.tensor([[2.0], [-1.0]]) + 0.5
    losses = []
    for i in range(60):
        out = sess.run((x, y))
        losses.append(float(out['loss']))
    assert out['global_step'] == 60
    assert losses[-1] < 0.05 * losses[0]
    sess.close()


def test_hybrid_sparse_model_single(monkeypatch, free_port):
    """An embedding+linear model under HYBRID: sparse rows update through
    the PS engine, dense through the optimizer."""
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)

    def build():
        model = nn.Module()
        model.emb = ParallaxEmbedding(50, 8)
        model.fc = nn.Linear(8, 2)
        opt = optim.Adagrad(
            list(model.parameters()), lr=0.3)

        def step_fn(m, batch):
            ids, y = batch
            out = m.fc(m.emb(ids)).mean(dim=0, keepdim=True)
            loss = ((out - y) ** 2).sum()
            return loss, {}

        return TrainGraph(model, opt, step_fn)

    sess, *_ = parallax_amd.parallel_run(
        build, None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    m = sess.graph.model
    w_before = m.emb.weight.detach().clone()
    ids = torch.tensor([1, 5, 5, 9])
    y = torch.tensor([[1.0, -1.0]])
    for _ in range(3):
        sess.run((ids, y))
    w_after = m.emb.weight.detach()
    touched = torch.tensor([1, 5, 9])
    untouched = torch.tensor([0, 2, 3, 49])
    assert not torch.allclose(w_before[touched], w_after[touched])
    assert torch.allclose(w_before[untouched], w_after[untouched])
    assert sess.engine.mode == 'HYBRID'
    assert sess.engine.sparse_engine is not None
    sess.close()


def test_degenerate_hybrid_falls_back_to_mpi(monkeypatch, free_port):
    """HYBRID + all-dense model -> MPI mode (reference runner.py:93-94)."""
    _worker_env(monkeypatch, free_port)
    sess, *_ = parallax_amd.parallel_run(
        lambda: build_simple_graph(), None,
        parallax_config=ParallaxConfig(run_option='HYBRID'))
    assert sess.engine.mode == 'MPI'
    sess.close()


def test_degenerate_hybrid_falls_back_to_ps(monkeypatch, free_port):
    """HYBRID + all-sparse model -> PS mode (reference runner.py:103-104)."""
    _worker_env(monkeypatch, free_port)

    def build():
        model = nn.Module()
        model.emb = ParallaxEmbedding(20, 4)
        opt = optim.SGD(model.parameters(), lr=0.1)

        def step_fn(m, batch):
            return m.emb(batch).sum(), {}

        return TrainGraph(model, opt, step_fn)

    sess, *_ = parallax_amd.parallel_run(
        build, None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    assert sess.engine.mode == 'PS'
    sess.close()


def test_ps_mode_mixed_model(monkeypatch, free_port):
    """Explicit run_option='PS': dense grads still go over the (fused)
    allreduce plane, sparse over the PS engine (doc/design.md divergence
    2); both engines active for a mixed model."""
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)

    def build():
        model = nn.Module()
        model.emb = ParallaxEmbedding(30, 4)
        model.fc = nn.Linear(4, 2)
        opt = optim.Adagrad(model.parameters(), lr=0.2)

        def step_fn(m, batch):
            ids, y = batch
            return ((m.fc(m.emb(ids)) - y) ** 2).mean(), {}

        return TrainGraph(model, opt, step_fn)

    sess, *_ = parallax_amd.parallel_run(
        build, None, parallax_config=ParallaxConfig(run_option='PS'))
    assert sess.engine.mode == 'PS'
    assert sess.engine.dense_engine is not None
    assert sess.engine.sparse_engine is not None
    out = sess.run((torch.tensor([1, 2, 2]), torch.randn(3, 2)))
    assert torch.isfinite(out['loss'])
    sess.close()


def test_profile_directory_scaffolding(tmp_path):
    """lib.create_profile_directory (reference lib.py:333-358): per-host/
    worker run_meta dirs + task_info manifest."""
    from parallax_amd.lib import create_profile_directory
    ri = {'worker': [{'hostname': 'h0', 'gpus': [0, 1]},
                     {'hostname': 'h1', 'gpus': [0]}]}
    create_profile_directory(str(tmp_path), ri)
    assert (tmp_path / 'task_info').exists()
    assert (tmp_path / 'h0' / 'worker:0' / 'run_meta').is_dir()
    assert (tmp_path / 'h1' / 'worker:1' / 'run_meta').is_dir()
    assert 'worker:0:h0:0,1' in (tmp_path / 'task_info').read_text()


def test_lm1b_tiny_loss_decreases(monkeypatch, free_port):
    """Tiny-config LM1B: the full hybrid stack (partitioned embeddings,
    sampled softmax, Adagrad, clipping) learns on a repeated batch."""
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)
    sess, *_ = parallax_amd.parallel_run(
        lambda: build_lm1b_graph(batch_size=4, num_steps=3, vocab_size=200,
                                 emb_dim=16, state_size=32, proj_size=16,
                                 num_sampled=32, min_partitions=4, lr=0.4),
        None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    assert sess.engine.mode == 'HYBRID'
    x = torch.randint(0, 200, (4, 3))
    y = torch.randint(0, 200, (4, 3))
    first = None
    state = None
    for i in range(30):
        out = sess.run((x, y, state))
        if first is None:
            first = float(out['loss'])
    assert float(out['loss']) < first
    sess.close()


def test_checkpoint_optimizer_state_and_shards(monkeypatch, free_port,
                                               tmp_path):
    """Checkpoint must round-trip the sparse optimizer state (Adagrad
    accumulators), the partitioned-table shard layout and the EMA shadow
    buffers — the reference saves ALL global variables (lib.py:38-56)."""
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)
    from parallax_amd.models.lm1b import build_lm1b_graph, synthetic_batch
    ck = CheckPointConfig(ckpt_dir=str(tmp_path / 'ck'), save_ckpt_steps=3)
    cfg = ParallaxConfig(run_option='HYBRID', ckpt_config=ck)
    kw = dict(batch_size=2, num_steps=2, vocab_size=100, emb_dim=8,
              state_size=16, proj_size=8, num_sampled=16,
              min_partitions=4, ema_decay=0.5)
    sess, *_ = parallax_amd.parallel_run(
        lambda: build_lm1b_graph(**kw), None, parallax_config=cfg)
    for _ in range(3):
        sess.run(synthetic_batch(2, 2, 100))
    m = sess.graph.model
    accum = sess.graph.optimizer.state[m.emb.weight]['sum'].clone()
    emb_w = m.emb.full_weight().clone()
    ema0 = m.ema_0.clone()
    sess.close()
    import torch.distributed as dist
    dist.destroy_process_group()

    payload = torch.load(str(tmp_path / 'ck' / 'model.ckpt-3.pt'),
                         weights_only=False)
    # shard layout on disk
    assert 'emb.parts.0' in payload['model']
    assert 'emb.parts.3' in payload['model']
    assert 'ema_0' in payload['model']

    monkeypatch.setenv('MASTER_PORT', str(free_port + 1))
    torch.manual_seed(99)  # different init; restore must overwrite
    sess2, *_ = parallax_amd.parallel_run(
        lambda: build_lm1b_graph(**kw), None, parallax_config=cfg)
    assert sess2.global_step == 3
    m2 = sess2.graph.model
    assert torch.allclose(m2.emb.full_weight(), emb_w)
    assert torch.allclose(m2.ema_0, ema0)
    accum2 = sess2.graph.optimizer.state[m2.emb.weight]['sum']
    assert torch.allclose(accum2, accum)
    # training continues
    out = sess2.run(synthetic_batch(2, 2, 100))
    assert torch.isfinite(out['loss'])
    sess2.close()


def test_checkpoint_roundtrip(monkeypatch, free_port, tmp_path):
    """Chief-only save on step cadence; restore resumes global_step and
    weights (reference §5.4 contract)."""
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)
    ck = CheckPointConfig(ckpt_dir=str(tmp_path / 'ckpt'),
                          save_ckpt_steps=5)
    cfg = ParallaxConfig(run_option='HYBRID', ckpt_config=ck)
    sess, *_ = parallax_amd.parallel_run(
        lambda: build_simple_graph(lr=0.05), None, parallax_config=cfg)
    x = torch.randn(8, 2)
    y = torch.randn(8, 1)
    for _ in range(7):
        sess.run((x, y))
    w_saved_run = sess.graph.model.w.detach().clone()
    sess.close()
    import torch.distributed as dist
    dist.destroy_process_group()

    # checkpoint exists and is keyed by user-visible names
    files = os.listdir(str(tmp_path / 'ckpt'))
    assert 'model.ckpt-5.pt' in files and 'checkpoint' in files
    payload = torch.load(str(tmp_path / 'ckpt' / 'model.ckpt-5.pt'),
                         weights_only=False)
    assert set(payload['model'].keys()) == {'w', 'b'}

    # new session restores at step 5
    monkeypatch.setenv('MASTER_PORT', str(free_port + 1))
    sess2, *_ = parallax_amd.parallel_run(
        lambda: build_simple_graph(lr=0.05), None, parallax_config=cfg)
    assert sess2.global_step == 5
    sess2.close()


def test_checkpoint_pure_bf16_masters_roundtrip(monkeypatch, free_port,
                                                tmp_path):
    """Pure-bf16 checkpoint: the fp32 masters ride the optimizer payload
    and must restore at fp32 (torch's default load would round them to
    the param dtype)."""
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)
    from parallax_amd.models.lm1b import build_lm1b_graph, synthetic_batch
    ck = CheckPointConfig(ckpt_dir=str(tmp_path / 'ck'), save_ckpt_steps=3)
    cfg = ParallaxConfig(run_option='HYBRID', ckpt_config=ck)
    kw = dict(batch_size=2, num_steps=2, vocab_size=100, emb_dim=8,
              state_size=16, proj_size=8, num_sampled=16,
              min_partitions=4, pure_bf16=True, ema_decay=None)
    sess, *_ = parallax_amd.parallel_run(
        lambda: build_lm1b_graph(**kw), None, parallax_config=cfg)
    for _ in range(3):
        sess.run(synthetic_batch(2, 2, 100))
    m = sess.graph.model
    master = sess.graph.optimizer.state[m.lstm.w]['master'].clone()
    w_bf16 = m.lstm.w.detach().clone()
    sess.close()
    import torch.distributed as dist
    dist.destroy_process_group()

    monkeypatch.setenv('MASTER_PORT', str(free_port + 1))
    torch.manual_seed(99)
    sess2, *_ = parallax_amd.parallel_run(
        lambda: build_lm1b_graph(**kw), None, parallax_config=cfg)
    assert sess2.global_step == 3
    m2 = sess2.graph.model
    assert m2.lstm.w.dtype == torch.bfloat16
    assert torch.equal(m2.lstm.w.detach(), w_bf16)
    master2 = sess2.graph.optimizer.state[m2.lstm.w]['master']
    assert master2.dtype == torch.float32
    assert torch.equal(master2, master)
    # the bf16 param is exactly the rounded master after restore + a step
    out = sess2.run(synthetic_batch(2, 2, 100))
    assert torch.isfinite(out['loss'])
    assert torch.equal(m2.lstm.w.detach(),
                       sess2.graph.optimizer.state[m2.lstm.w]['master']
                       .to(torch.bfloat16))
    sess2.close()


def test_dense_engine_overlap_toggle(monkeypatch, free_port):
    """set_overlap(False/True) swaps hook-driven launches on and off (the
    hipGraph-capture path turns overlap off because collective launches
    move after replay)."""
    _worker_env(monkeypatch, free_port)
    sess, *_ = parallax_amd.parallel_run(
        lambda: build_simple_graph(), None,
        parallax_config=ParallaxConfig(run_option='MPI'))
    eng = sess.engine.dense_engine
    assert eng is not None
    # world_size == 1: overlap stays off regardless (nothing to launch)
    assert eng.overlap is False
    eng.set_overlap(True)
    assert eng.overlap is False and eng._hooks == []
    # bucket bookkeeping is deterministic: every param assigned once
    n_params = sum(len(b.params) for b in eng._buckets)
    assert n_params == len(eng.infos)
    assert set(eng._param_bucket) == {id(gi.param) for gi in eng.infos}
    sess.close()


def test_dense_grad_on_sparse_param(monkeypatch, free_port):
    """A sparse-tagged table used in a DENSE matmul (tied full-softmax
    eval path) produces a dense grad; the capture hook treats every row
    as touched and the engine still applies it (sparse_engine.py
    _capture_hook dense branch)."""
    import torch
    import torch.nn as nn
    import parallax_amd
    from parallax_amd import optim
    from parallax_amd.config import ParallaxConfig
    from parallax_amd.session import TrainGraph
    from parallax_amd.ops.embedding import ParallaxEmbedding

    monkeypatch.setenv('WORLD_SIZE', '1')
    monkeypatch.setenv('RANK', '0')
    monkeypatch.setenv('MASTER_ADDR', '127.0.0.1')
    monkeypatch.setenv('MASTER_PORT', str(free_port))
    torch.manual_seed(9)

    def build():
        model = nn.Module()
        model.emb = ParallaxEmbedding(20, 6)
        model.fc = nn.Linear(6, 6)
        opt = optim.Adagrad(model.parameters(), lr=0.3)

        def step_fn(m, batch):
            ids, y = batch
            h = m.fc(m.emb(ids))                   # sparse use
            logits = h @ m.emb.weight.t()          # DENSE use of the table
            loss = torch.nn.functional.cross_entropy(logits, y)
            return loss, {}

        return TrainGraph(model, opt, step_fn)

    sess, *_ = parallax_amd.parallel_run(
        build, None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    w0 = sess.graph.model.emb.weight.detach().clone()
    ids = torch.tensor([1, 3, 3, 7])
    y = torch.tensor([0, 1, 2, 3])
    first = None
    for _ in range(6):
        out = sess.run((ids, y))
        if first is None:
            first = float(out['loss'])
    assert float(out['loss']) < first
    w1 = sess.graph.model.emb.weight.detach()
    # the dense use touches EVERY row: untouched-by-lookup rows moved too
    assert not torch.allclose(w0[15], w1[15])
    sess.close()
