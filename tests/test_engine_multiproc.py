"""Multi-process CPU tests (gloo backend, world_size=2): the distributed
paths the driver exercises on the 8-GPU node, validated here without a GPU
(BASELINE config 1 class of tests).  Rendezvous over 127.0.0.1."""

import os

import pytest
import torch
import torch.multiprocessing as mp

from parallax_amd.lib import get_free_port

WORLD = 2


def _init_env(rank, world, port):
    os.environ['RANK'] = str(rank)
    os.environ['LOCAL_RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)


def _spawn_once(fn, port, world, timeout):
    ctx = mp.get_context('spawn')
    procs = []
    err_q = ctx.SimpleQueue()
    for r in range(world):
        p = ctx.Process(target=_child_main, args=(fn, r, world, port, err_q))
        p.start()
        procs.append(p)
    for p in procs:
        p.join(timeout)
    errs = []
    while not err_q.empty():
        errs.append(err_q.get())
    infra_only = not errs   # no Python traceback from any worker
    for p in procs:
        if p.is_alive():
            p.terminate()
            errs.append('worker timed out')
            infra_only = False
        elif p.exitcode != 0:
            errs.append('worker exit %s' % p.exitcode)
    return errs, infra_only


_INFRA_SIGNATURES = ('Connection refused', 'Address already in use',
                     'Connection reset', 'Socket Timeout', 'EOFError',
                     'Connection closed by peer', 'Broken pipe',
                     'timed out')


def _infra_errs(errs):
    """True when every error is an infrastructure signature (rendezvous/
    transport), not a test assertion."""
    return errs and all(
        not isinstance(e, str) or not e.startswith('worker timed out')
        for e in errs) and all(
        (isinstance(e, str) and e.startswith('worker exit'))
        or any(sig in str(e) for sig in _INFRA_SIGNATURES)
        for e in errs)


def _spawn(fn, port, world=WORLD, timeout=180):
    errs, infra_only = _spawn_once(fn, port, world, timeout)
    if errs and (infra_only or _infra_errs(errs)):
        # a worker died without raising, or raised a pure transport error
        # (rendezvous port clash / gloo abort): one retry on a fresh
        # port; genuine test failures (assertion tracebacks) never retry
        errs, _ = _spawn_once(fn, get_free_port(), world, timeout)
    assert not errs, '\n'.join(str(e) for e in errs)


def _child_main(fn, rank, world, port, err_q):
    try:
        _init_env(rank, world, port)
        fn(rank, world)
    except Exception:
        import traceback
        err_q.put(traceback.format_exc())
        raise
    finally:
        import torch.distributed as dist
        if dist.is_initialized():
            dist.destroy_process_group()


# ---------------------------------------------------------------------------
# child bodies (module-level: picklable under spawn)
# ---------------------------------------------------------------------------

def _body_allreduce_engine(rank, world):
    import torch.nn as nn
    from parallax_amd.parallel.comm import CommContext
    from parallax_amd.parallel.dense_engine import DenseAllReduceEngine
    from parallax_amd.tagging import classify_params

    comm = CommContext()
    torch.manual_seed(123)  # identical init on both ranks
    model = nn.Sequential(nn.Linear(4, 8), nn.Linear(8, 2))
    infos = classify_params(model)
    eng = DenseAllReduceEngine(comm, infos, bucket_bytes=64,  # force >1 bucket
                               overlap=True)
    eng.prepare_step()
    torch.manual_seed(500 + rank)  # different data per rank
    x = torch.randn(6, 4)
    loss = model(x).pow(2).sum()
    loss.backward()
    eng.reduce_gradients()
    # grads must now equal the average over both ranks: recompute both
    grads_mine = [p.grad.clone() for p in model.parameters()]
    ref_model = nn.Sequential(nn.Linear(4, 8), nn.Linear(8, 2))
    ref_model.load_state_dict(model.state_dict())
    expected = [torch.zeros_like(p) for p in ref_model.parameters()]
    for r in range(world):
        ref_model.zero_grad()
        torch.manual_seed(500 + r)
        xr = torch.randn(6, 4)
        ref_model(xr).pow(2).sum().backward()
        for e, p in zip(expected, ref_model.parameters()):
            e += p.grad / world
    for g, e in zip(grads_mine, expected):
        assert torch.allclose(g, e, atol=1e-5), (g - e).abs().max()
    eng.close()


def _body_sparse_allgather(rank, world):
    import torch.nn as nn
    from parallax_amd import optim
    from parallax_amd.parallel.comm import CommContext
    from parallax_amd.parallel.sparse_engine import SparseAllGatherEngine
    from parallax_amd.ops.embedding import ParallaxEmbedding
    from parallax_amd.tagging import classify_params

    comm = CommContext()
    torch.manual_seed(7)
    model = nn.Module()
    model.emb = ParallaxEmbedding(30, 4)
    opt = optim.Adagrad(model.parameters(), lr=0.5)
    infos = classify_params(model)
    eng = SparseAllGatherEngine(comm, infos, opt)
    eng.prepare_step()
    ids = torch.tensor([1 + rank, 5])  # rank0: {1,5}, rank1: {2,5}
    out = model.emb(ids)
    out.sum().backward()
    eng.reduce_and_apply()
    # both ranks must hold IDENTICAL tables afterwards
    w = model.emb.weight.detach()
    ws = [torch.zeros_like(w) for _ in range(world)]
    torch.distributed.all_gather(ws, w)
    assert torch.allclose(ws[0], ws[1], atol=1e-6)
    # rows 1,2,5 updated; row 5 got grad from both ranks (summed)
    torch.manual_seed(7)
    ref = ParallaxEmbedding(30, 4)
    assert not torch.allclose(w[5], ref.weight[5])
    assert not torch.allclose(w[1], ref.weight[1])
    assert not torch.allclose(w[2], ref.weight[2])
    assert torch.allclose(w[9], ref.weight[9])
    eng.close()


def _body_sparse_ps(rank, world):
    import torch.nn as nn
    from parallax_amd import optim
    from parallax_amd.parallel.comm import CommContext
    from parallax_amd.parallel.sparse_engine import SparsePSEngine
    from parallax_amd.ops.embedding import PartitionedEmbedding
    from parallax_amd.partitions import FixedSizePartitioner
    from parallax_amd.tagging import classify_params

    comm = CommContext()
    torch.manual_seed(11)
    model = nn.Module()
    model.emb = PartitionedEmbedding(40, 4, FixedSizePartitioner(4))
    opt = optim.Adagrad(model.parameters(), lr=0.5)
    infos = classify_params(model)
    assert infos[0].partition == (4, (0, 10, 20, 30))
    eng = SparsePSEngine(comm, infos, opt, local_aggregation=True)
    eng.prepare_step()
    ids = torch.tensor([0, 11 + rank, 25, 25])  # dup row tests local agg
    out = model.emb(ids)
    out.sum().backward()
    eng.reduce_and_apply()
    # replicated tables must be identical across ranks after apply
    full = model.emb.full_weight()
    fulls = [torch.zeros_like(full) for _ in range(world)]
    torch.distributed.all_gather(fulls, full)
    assert torch.allclose(fulls[0], fulls[1], atol=1e-6), \
        (fulls[0] - fulls[1]).abs().max()
    # updated rows: 0 (both ranks), 11, 12, 25
    torch.manual_seed(11)
    ref = PartitionedEmbedding(40, 4, FixedSizePartitioner(4))
    ref_full = ref.full_weight()
    for row in (0, 11, 12, 25):
        assert not torch.allclose(full[row], ref_full[row]), row
    for row in (3, 20, 39):
        assert torch.allclose(full[row], ref_full[row]), row
    # local aggregation semantics: the duplicated row 25 must have been
    # summed BEFORE the adagrad update (one update with grad 2g, not two
    # with g) — compare against a single-process reference
    o_ref = optim.Adagrad(ref.parameters(), lr=0.5)
    ref_ids = torch.tensor([0, 0, 11, 12, 25, 25, 25, 25])
    ref(ref_ids).sum().backward()
    g = ref.weight.grad.coalesce()
    o_ref.sparse_row_step(ref.weight, g.indices().flatten(), g.values())
    assert torch.allclose(full, ref.full_weight(), atol=1e-6)
    eng.close()


def _body_parallel_run_hybrid(rank, world):
    """Full parallel_run worker path with a mixed dense+sparse model in
    HYBRID mode; checks loss decreases and models stay in sync."""
    import torch.nn as nn
    import parallax_amd
    from parallax_amd import optim
    from parallax_amd.config import ParallaxConfig
    from parallax_amd.session import TrainGraph
    from parallax_amd.ops.embedding import PartitionedEmbedding
    from parallax_amd.partitions import FixedSizePartitioner

    def build():
        torch.manual_seed(21 + rank)  # broadcast must fix divergent init
        model = nn.Module()
        model.emb = PartitionedEmbedding(40, 8, FixedSizePartitioner(4))
        model.fc = nn.Linear(8, 2)
        opt = optim.Adagrad(model.parameters(), lr=0.3)

        def step_fn(m, batch):
            ids, y = batch
            out = m.fc(m.emb(ids))
            loss = ((out - y) ** 2).mean()
            return loss, {}

        return TrainGraph(model, opt, step_fn)

    sess, num_workers, worker_id, n_rep = parallax_amd.parallel_run(
        build, None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    assert num_workers == world and worker_id == rank
    torch.manual_seed(100 + rank)
    ids = torch.randint(0, 40, (16,))
    y = torch.randn(16, 2) * 0.1
    first = None
    for i in range(20):
        out = sess.run((ids, y))
        if first is None:
            first = float(out['loss'])
    assert float(out['loss']) < first
    # dense params must be identical across ranks (same broadcast init +
    # same averaged grads)
    w = sess.graph.model.fc.weight.detach()
    ws = [torch.zeros_like(w) for _ in range(world)]
    torch.distributed.all_gather(ws, w)
    assert torch.allclose(ws[0], ws[1], atol=1e-5)
    # sparse mirrors consistent
    full = sess.graph.model.emb.full_weight()
    fs = [torch.zeros_like(full) for _ in range(world)]
    torch.distributed.all_gather(fs, full)
    assert torch.allclose(fs[0], fs[1], atol=1e-5)
    sess.close()


def _body_comm_collectives(rank, world):
    from parallax_amd.parallel.comm import CommContext
    comm = CommContext()
    # all_gather_v with different lengths
    t = torch.arange(3 + rank, dtype=torch.float32).reshape(-1, 1)
    outs = comm.all_gather_v(t)
    assert [o.shape[0] for o in outs] == [3, 4]
    assert outs[1][3, 0] == 3.0
    # all_to_all_v
    sends = [torch.full((rank + 1, 2), float(10 * rank + d))
             for d in range(world)]
    recv = comm.all_to_all_v(sends)
    # from rank r we receive (r+1) rows filled with 10*r + my_rank
    for r in range(world):
        assert recv[r].shape == (r + 1, 2)
        assert torch.all(recv[r] == 10 * r + rank)
    # exchange_rows (the N7 (ids, rows) convenience)
    ids = [torch.arange(rank + 1) + 100 * rank for _ in range(world)]
    rows = [torch.full((rank + 1, 3), float(rank)) for _ in range(world)]
    rids, rrows = comm.exchange_rows(ids, rows)
    for r in range(world):
        assert rids[r].shape[0] == r + 1
        assert torch.all(rrows[r] == float(r))
    # all_gather_v_multi single counts exchange
    multi = comm.all_gather_v_multi([t, t + 1.0])
    assert torch.allclose(torch.cat(multi[1][0:1])[0],
                          torch.cat(multi[0][0:1])[0] + 1.0)
    # all_to_all_v with ZERO-length messages (rank 0 sends nothing to
    # anyone) — the isend/irecv fallback must not post empty sends
    zsend = [torch.zeros((0 if rank == 0 else 2, 3)) + rank
             for _ in range(world)]
    zrecv = comm.all_to_all_v(zsend)
    for r in range(world):
        assert zrecv[r].shape[0] == (0 if r == 0 else 2)
        if zrecv[r].shape[0]:
            assert torch.all(zrecv[r] == r)
    # all_gather_v where one rank contributes nothing
    e = torch.ones(0 if rank == 0 else 5, 2)
    eouts = comm.all_gather_v(e)
    assert [o.shape[0] for o in eouts] == [0 if r == 0 else 5
                                           for r in range(world)]


def _body_average_sparse(rank, world):
    """average_sparse=True: the update uses the occurrence-count average
    of duplicate rows ACROSS ranks (reference average_option
    SPARSE_AVERAGE_BY_COUNTER, graph_transform_lib.py:385-390) — counts
    must survive the local pre-aggregation."""
    import torch.nn as nn
    from parallax_amd import optim
    from parallax_amd.parallel.comm import CommContext
    from parallax_amd.parallel.sparse_engine import SparsePSEngine
    from parallax_amd.ops.embedding import ParallaxEmbedding
    from parallax_amd.tagging import classify_params

    comm = CommContext()
    torch.manual_seed(77)
    model = nn.Module()
    model.emb = ParallaxEmbedding(20, 4)
    opt = optim.SGD(model.parameters(), lr=1.0)
    eng = SparsePSEngine(comm, classify_params(model), opt,
                         average_sparse=True, local_aggregation=True)
    eng.prepare_step()
    # rank 0 uses row 5 three times, rank 1 once -> avg divisor 4
    n = 3 if rank == 0 else 1
    ids = torch.full((n,), 5, dtype=torch.long)
    before = model.emb.weight[5].detach().clone()
    out = model.emb(ids)
    out.sum().backward()
    eng.reduce_and_apply()
    after = model.emb.weight[5].detach()
    # each use contributes grad of ones; sum = 4*ones, count = 4 ->
    # averaged grad = ones; SGD lr=1 -> delta = -1
    assert torch.allclose(after - before, -torch.ones(4), atol=1e-5), \
        (after - before)
    eng.close()


def _body_lm1b_tiny_full_stack(rank, world, pure_bf16=False):
    """Tiny LM1B through the COMPLETE per-step protocol on several ranks:
    partitioned tables, sampled softmax, clip, EMA, dense AR, sparse PS —
    the collective-order rehearsal for the 8-GPU node.  pure_bf16
    additionally rehearses bf16 dense grads over the wire (gloo here,
    RCCL on the node) with fp32 masters."""
    import parallax_amd
    from parallax_amd.config import ParallaxConfig
    from parallax_amd.models import lm1b as M

    sess, num_workers, worker_id, _ = parallax_amd.parallel_run(
        lambda: M.build_lm1b_graph(batch_size=4, num_steps=3,
                                   vocab_size=300, emb_dim=16,
                                   state_size=32, proj_size=16,
                                   num_sampled=32, min_partitions=4,
                                   lr=0.3, pure_bf16=pure_bf16),
        None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    state = None
    for i in range(8):
        out = sess.run(M.synthetic_batch(4, 3, 300, state=state))
        state = out.get('state')
        assert float(out['loss']) == float(out['loss'])
    # all replicated state identical across ranks at the end
    m = sess.graph.model
    for t in (m.emb.weight.detach(), m.softmax_b.detach(),
              m.lstm.w.detach(), m.ema_0.detach()):
        ts = [torch.zeros_like(t) for _ in range(world)]
        torch.distributed.all_gather(ts, t)
        assert torch.allclose(ts[0].float(), ts[-1].float(), atol=1e-5)
    sess.close()


def _body_lm1b_pure_bf16(rank, world):
    _body_lm1b_tiny_full_stack(rank, world, pure_bf16=True)


def _run_lm1b_steps(cfg, batches, seed=0):
    """Build tiny LM1B under cfg, run the given batches, return the final
    full tables (collective in sharded mode) + loss trace."""
    import parallax_amd
    from parallax_amd.models import lm1b as M
    torch.manual_seed(seed)
    sess, *_ = parallax_amd.parallel_run(
        lambda: M.build_lm1b_graph(batch_size=4, num_steps=3,
                                   vocab_size=300, emb_dim=16,
                                   state_size=32, proj_size=16,
                                   num_sampled=32, min_partitions=4,
                                   lr=0.3, ema_decay=None),
        None, parallax_config=cfg)
    losses = []
    for b in batches:
        out = sess.run((b[0], b[1], None))
        losses.append(float(out['loss']))
    m = sess.graph.model
    emb = m.emb.full_weight().clone()
    sw = m.softmax_w.full_weight().clone()
    sess.close()
    return emb, sw, losses


def _body_sharded_tables_equivalence(rank, world, run_option='HYBRID',
                                     average_sparse=False):
    """replicate_variables=False: shard-only storage must produce the SAME
    trained tables as the replicated engine on identical data (the
    reference's PS placement is a storage choice, not a semantics
    change).  Checked for both sum and count-average sparse semantics."""
    import torch.distributed as dist
    from parallax_amd.config import ParallaxConfig

    gen = torch.Generator().manual_seed(777)  # identical on every rank
    batches = [(torch.randint(0, 300, (4, 3), generator=gen),
                torch.randint(0, 300, (4, 3), generator=gen))
               for _ in range(6)]

    cfg_sh = ParallaxConfig(run_option=run_option,
                            average_sparse=average_sparse)
    cfg_sh.communication_config.ps_config.replicate_variables = False
    emb_s, sw_s, losses_s = _run_lm1b_steps(cfg_sh, batches)
    # full_weight() gathers identically on every rank
    g = [torch.zeros_like(emb_s) for _ in range(world)]
    dist.all_gather(g, emb_s)
    assert torch.allclose(g[0], g[-1], atol=1e-6)

    dist.destroy_process_group()
    os.environ['MASTER_PORT'] = str(int(os.environ['MASTER_PORT']) + 1)
    cfg_rep = ParallaxConfig(run_option=run_option,
                             average_sparse=average_sparse)
    emb_r, sw_r, losses_r = _run_lm1b_steps(cfg_rep, batches)

    assert torch.allclose(emb_s, emb_r, atol=1e-4), \
        (emb_s - emb_r).abs().max()
    assert torch.allclose(sw_s, sw_r, atol=1e-4)
    for a, b in zip(losses_s, losses_r):
        assert abs(a - b) < 1e-3, (a, b)


def _body_sharded_ps_average(rank, world):
    _body_sharded_tables_equivalence(rank, world, run_option='PS',
                                     average_sparse=True)


def _body_sharded_uneven_partitions(rank, world):
    """to_sharded with a partition count NOT divisible by world (5 parts,
    world 2 -> rank 0 owns {0,2,4}, rank 1 owns {1,3}): the shard slicing,
    local indexing and full_weight reassembly must stay exact."""
    import torch.distributed as dist
    from parallax_amd.parallel.comm import CommContext
    from parallax_amd.ops.embedding import PartitionedEmbedding
    from parallax_amd.partitions import FixedSizePartitioner

    comm = CommContext()
    torch.manual_seed(7)  # same init everywhere (no broadcast needed)
    emb = PartitionedEmbedding(103, 8, FixedSizePartitioner(5))
    ref = emb.weight.detach().clone()
    emb.to_sharded(comm)
    owned_rows = sum(emb.sizes[k] for k in emb._owned)
    assert emb.weight.shape == (owned_rows, 8)
    # local_index maps a global id of an owned partition to its local row
    for k in emb._owned:
        gid = torch.tensor([emb.offsets[k]])
        lid = emb.local_index(gid)
        assert torch.equal(emb.weight.detach()[lid], ref[gid])
    # collective reassembly returns the exact original table
    full = emb.full_weight()
    assert torch.equal(full, ref)
    # a sharded forward fetches the right rows
    ids = torch.tensor([[0, 50, 102, 17]])
    out = emb(ids)
    assert torch.allclose(out[0], ref[ids[0]])
    dist.barrier()


def _body_sharded_checkpoint(rank, world):
    """Sharded-mode checkpoint: collective parts.k save (replicated
    layout on disk), per-rank optimizer sidecars, restore resumes."""
    import tempfile
    import parallax_amd
    import torch.distributed as dist
    from parallax_amd.config import ParallaxConfig, CheckPointConfig
    from parallax_amd.models import lm1b as M

    tmp = os.path.join(tempfile.gettempdir(),
                       'px_shard_ck_%s' % os.environ['MASTER_PORT'])
    cfg = ParallaxConfig(
        run_option='HYBRID',
        ckpt_config=CheckPointConfig(ckpt_dir=tmp, save_ckpt_steps=2))
    cfg.communication_config.ps_config.replicate_variables = False
    kw = dict(batch_size=4, num_steps=3, vocab_size=300, emb_dim=16,
              state_size=32, proj_size=16, num_sampled=32,
              min_partitions=4, lr=0.3, ema_decay=None)
    torch.manual_seed(0)
    sess, *_ = parallax_amd.parallel_run(
        lambda: M.build_lm1b_graph(**kw), None, parallax_config=cfg)
    for _ in range(2):
        sess.run(M.synthetic_batch(4, 3, 300))
    emb_saved = sess.graph.model.emb.full_weight().clone()
    sess.close()
    dist.barrier()
    if rank == 0:
        import glob as _g
        assert os.path.isfile(os.path.join(tmp, 'model.ckpt-2.pt'))
        sides = _g.glob(os.path.join(tmp, '*.optshard*-of-%d.pt' % world))
        assert len(sides) == world, sides
        payload = torch.load(os.path.join(tmp, 'model.ckpt-2.pt'),
                             weights_only=False)
        assert payload['sharded_world'] == world
        assert 'emb.parts.0' in payload['model']
    dist.destroy_process_group()

    os.environ['MASTER_PORT'] = str(int(os.environ['MASTER_PORT']) + 1)
    torch.manual_seed(99)  # different init; restore must overwrite
    sess2, *_ = parallax_amd.parallel_run(
        lambda: M.build_lm1b_graph(**kw), None, parallax_config=cfg)
    assert sess2.global_step == 2
    m2 = sess2.graph.model
    assert torch.allclose(m2.emb.full_weight(), emb_saved, atol=1e-6)
    # accumulators restored from this rank's sidecar
    st = sess2.graph.optimizer.state.get(m2.emb.weight)
    assert st is not None and 'sum' in st
    out = sess2.run(M.synthetic_batch(4, 3, 300))
    assert torch.isfinite(out['loss'])
    sess2.close()


# ---------------------------------------------------------------------------

@pytest.mark.parametrize('body', [
    _body_comm_collectives,
    _body_allreduce_engine,
    _body_sparse_allgather,
    _body_sparse_ps,
    _body_parallel_run_hybrid,
], ids=['collectives', 'dense_allreduce', 'sparse_allgather', 'sparse_ps',
        'parallel_run_hybrid'])
def test_multiproc(body):
    _spawn(body, get_free_port())


def test_lm1b_full_stack_world4():
    _spawn(_body_lm1b_tiny_full_stack, get_free_port(), world=4,
           timeout=300)


def test_lm1b_pure_bf16_world2():
    _spawn(_body_lm1b_pure_bf16, get_free_port(), world=2, timeout=300)


def test_sharded_tables_equivalence_world2():
    _spawn(_body_sharded_tables_equivalence, get_free_port(), world=2,
           timeout=300)


def test_sharded_checkpoint_world2():
    import shutil
    import tempfile
    port = get_free_port()
    # the body keys its ckpt dir by MASTER_PORT; free ports recycle, so a
    # stale dir from an earlier test would be restored as a checkpoint
    stale = os.path.join(tempfile.gettempdir(), 'px_shard_ck_%d' % port)
    shutil.rmtree(stale, ignore_errors=True)
    _spawn(_body_sharded_checkpoint, port, world=2, timeout=300)


def test_sharded_ps_average_world2():
    _spawn(_body_sharded_ps_average, get_free_port(), world=2,
           timeout=300)


def test_sharded_uneven_partitions_world2():
    _spawn(_body_sharded_uneven_partitions, get_free_port(), world=2,
           timeout=180)


def _body_sharded_save_for_crossmode(rank, world):
    """Save a sharded checkpoint + the expected full table for the
    cross-mode restore check in the parent process."""
    import parallax_amd
    import torch.distributed as dist
    from parallax_amd.config import ParallaxConfig, CheckPointConfig
    from parallax_amd.models import lm1b as M
    tmp = os.environ['PX_TEST_CKPT_DIR']
    cfg = ParallaxConfig(
        run_option='HYBRID',
        ckpt_config=CheckPointConfig(ckpt_dir=tmp, save_ckpt_steps=2))
    cfg.communication_config.ps_config.replicate_variables = False
    torch.manual_seed(0)
    sess, *_ = parallax_amd.parallel_run(
        lambda: M.build_lm1b_graph(
            batch_size=4, num_steps=3, vocab_size=300, emb_dim=16,
            state_size=32, proj_size=16, num_sampled=32, min_partitions=4,
            lr=0.3, ema_decay=None),
        None, parallax_config=cfg)
    for _ in range(2):
        sess.run(M.synthetic_batch(4, 3, 300))
    full = sess.graph.model.emb.full_weight()
    # expected FULL Adagrad accumulator (each row owned by exactly one
    # rank; zeros elsewhere + all_reduce SUM assembles it)
    mod = sess.graph.model.emb
    st = sess.graph.optimizer.state[mod.weight]['sum']
    full_sum = torch.zeros(mod.num_embeddings, mod.embedding_dim)
    local_off = 0
    for k in mod._owned:
        off, size = mod.offsets[k], mod.sizes[k]
        full_sum[off:off + size] = st[local_off:local_off + size]
        local_off += size
    dist.all_reduce(full_sum)
    if rank == 0:
        torch.save(full, os.path.join(tmp, 'expected_emb.pt'))
        torch.save(full_sum, os.path.join(tmp, 'expected_sum.pt'))
    sess.close()
    dist.barrier()


def test_sharded_checkpoint_restores_replicated(monkeypatch, tmp_path):
    """A shard-only run's checkpoint keeps the replicated parts.k layout,
    so it restores into a plain (replicated, world-1) session."""
    import parallax_amd
    from parallax_amd.config import ParallaxConfig, CheckPointConfig
    from parallax_amd.models import lm1b as M
    ckdir = str(tmp_path / 'ck')
    os.environ['PX_TEST_CKPT_DIR'] = ckdir
    try:
        _spawn(_body_sharded_save_for_crossmode, get_free_port(), world=2,
               timeout=300)
    finally:
        os.environ.pop('PX_TEST_CKPT_DIR', None)
    expected = torch.load(os.path.join(ckdir, 'expected_emb.pt'),
                          weights_only=False)

    monkeypatch.setenv('WORLD_SIZE', '1')
    monkeypatch.setenv('RANK', '0')
    monkeypatch.setenv('MASTER_ADDR', '127.0.0.1')
    monkeypatch.setenv('MASTER_PORT', str(get_free_port()))
    cfg = ParallaxConfig(
        run_option='HYBRID',
        ckpt_config=CheckPointConfig(ckpt_dir=ckdir, save_ckpt_steps=100))
    torch.manual_seed(99)  # different init; restore must overwrite
    sess, *_ = parallax_amd.parallel_run(
        lambda: M.build_lm1b_graph(
            batch_size=4, num_steps=3, vocab_size=300, emb_dim=16,
            state_size=32, proj_size=16, num_sampled=32, min_partitions=4,
            lr=0.3, ema_decay=None),
        None, parallax_config=cfg)
    assert sess.global_step == 2
    assert torch.allclose(sess.graph.model.emb.full_weight(), expected)
    # per-row optimizer state RESHARDED from the world-2 sidecars into
    # this replicated world-1 session (layout manifests in the sidecars)
    expected_sum = torch.load(os.path.join(ckdir, 'expected_sum.pt'),
                              weights_only=False)
    st = sess.graph.optimizer.state.get(sess.graph.model.emb.weight)
    assert st is not None and 'sum' in st, 'optimizer state not resharded'
    assert st['sum'].shape == expected_sum.shape
    assert torch.allclose(st['sum'], expected_sum, atol=1e-5)
    out = sess.run(M.synthetic_batch(4, 3, 300))
    assert torch.isfinite(out['loss'])
    sess.close()
    import torch.distributed as dist
    dist.destroy_process_group()


def test_average_sparse_world2():
    _spawn(_body_average_sparse, get_free_port())


def _body_unused_dense_param(rank, world):
    """A conditionally-unused dense parameter must not wedge or corrupt
    the bucketed all-reduce (DDP-style zero-contribution semantics; the
    reference never hits this — TF graphs compute every registered grad —
    so this is robustness for eager user models with branches)."""
    import torch.nn as nn
    from parallax_amd.parallel.comm import CommContext
    from parallax_amd.parallel.dense_engine import DenseAllReduceEngine
    from parallax_amd.tagging import classify_params

    comm = CommContext()
    torch.manual_seed(321)
    model = nn.Module()
    model.a = nn.Linear(4, 4)
    model.b = nn.Linear(4, 4)   # used only on rank 0 / only on odd steps
    model.c = nn.Linear(4, 2)
    infos = classify_params(model)
    eng = DenseAllReduceEngine(comm, infos, bucket_bytes=64,
                               overlap=True)
    for step in range(3):
        eng.prepare_step()
        model.zero_grad(set_to_none=True)
        torch.manual_seed(1000 * step + rank)
        x = torch.randn(5, 4)
        h = model.a(x)
        use_b = (rank == 0) if step == 0 else (step % 2 == 1)
        if use_b:
            h = model.b(h)
        loss = model.c(h).pow(2).sum()
        loss.backward()
        eng.reduce_gradients()
        # every param now has a grad, identical across ranks
        for p in model.parameters():
            assert p.grad is not None
            got = [torch.zeros_like(p.grad) for _ in range(world)]
            torch.distributed.all_gather(got, p.grad)
            assert torch.allclose(got[0], got[1], atol=1e-6)
        # recompute the expected average by hand on every rank
        ref = nn.Module()
        ref.a, ref.b, ref.c = (nn.Linear(4, 4), nn.Linear(4, 4),
                               nn.Linear(4, 2))
        ref.load_state_dict(model.state_dict())
        expected = {n: torch.zeros_like(p)
                    for n, p in ref.named_parameters()}
        for r in range(world):
            ref.zero_grad()
            torch.manual_seed(1000 * step + r)
            xr = torch.randn(5, 4)
            hr = ref.a(xr)
            if (r == 0) if step == 0 else (step % 2 == 1):
                hr = ref.b(hr)
            ref.c(hr).pow(2).sum().backward()
            for n, p in ref.named_parameters():
                if p.grad is not None:
                    expected[n] += p.grad / world
        for (n, p) in model.named_parameters():
            assert torch.allclose(p.grad, expected[n], atol=1e-5), \
                (n, (p.grad - expected[n]).abs().max())
    eng.close()


def test_unused_dense_param_world2():
    _spawn(_body_unused_dense_param, get_free_port())


def _body_mpi_average_world(rank, world):
    """MPI mode + average_sparse: divide by WORLD SIZE (Horovod
    hvd.allreduce(IndexedSlices, average=True) semantics,
    mpi/graph_transform.py:51-53) — not by occurrence count."""
    import torch.nn as nn
    from parallax_amd import optim
    from parallax_amd.parallel.comm import CommContext
    from parallax_amd.parallel.sparse_engine import SparseAllGatherEngine
    from parallax_amd.ops.embedding import ParallaxEmbedding
    from parallax_amd.tagging import classify_params

    comm = CommContext()
    torch.manual_seed(77)
    model = nn.Module()
    model.emb = ParallaxEmbedding(20, 4)
    opt = optim.SGD(model.parameters(), lr=1.0)
    eng = SparseAllGatherEngine(comm, classify_params(model), opt,
                                average_sparse=True)
    eng.prepare_step()
    # rank 0 uses row 5 three times, rank 1 once -> sum = 4*ones,
    # Horovod average divides by world (2) -> grad = 2*ones -> delta = -2
    n = 3 if rank == 0 else 1
    ids = torch.full((n,), 5, dtype=torch.long)
    before = model.emb.weight[5].detach().clone()
    model.emb(ids).sum().backward()
    eng.reduce_and_apply()
    after = model.emb.weight[5].detach()
    assert torch.allclose(after - before, -2.0 * torch.ones(4),
                          atol=1e-5), (after - before)
    eng.close()


def test_mpi_average_sparse_world2():
    _spawn(_body_mpi_average_world, get_free_port())


def _body_padded_exchange_equivalence(rank, world):
    """World-2 padded (fixed-shape, sync-free) sparse exchange ==
    dynamic-size exchange, for PS (count-average) and MPI (world-average)
    engines."""
    import torch.nn as nn
    from parallax_amd import optim
    from parallax_amd.parallel.comm import CommContext
    from parallax_amd.parallel.sparse_engine import (SparsePSEngine,
                                                     SparseAllGatherEngine)
    from parallax_amd.ops.embedding import ParallaxEmbedding
    from parallax_amd.tagging import classify_params

    comm = CommContext()
    for cls, avg in [(SparsePSEngine, True), (SparsePSEngine, False),
                     (SparseAllGatherEngine, True)]:
        results = {}
        for mode in ('force', '0'):
            os.environ['PARALLAX_PADDED_SPARSE'] = mode
            try:
                torch.manual_seed(13)  # same init both modes and ranks
                model = nn.Module()
                model.emb = ParallaxEmbedding(50, 4)
                opt = optim.Adagrad(model.parameters(), lr=0.5)
                eng = cls(comm, classify_params(model), opt,
                          average_sparse=avg)
                eng.prepare_step()
                # same COUNT per rank (static shapes), different ids
                ids = torch.tensor([1 + rank, 5, 5, 30 + 2 * rank])
                model.emb(ids).sum().backward()
                eng.reduce_and_apply()
                results[mode] = model.emb.weight.detach().clone()
                if mode == 'force':
                    assert eng._padded and eng._counts_static is True
                eng.close()
            finally:
                os.environ.pop('PARALLAX_PADDED_SPARSE', None)
        assert torch.allclose(results['force'], results['0'], atol=1e-5), \
            (cls.__name__, avg,
             (results['force'] - results['0']).abs().max())
    # dynamic fallback: DIFFERENT per-rank counts must be detected once
    # and the engine must still produce identical tables on both ranks
    os.environ['PARALLAX_PADDED_SPARSE'] = 'force'
    try:
        torch.manual_seed(29)
        model = nn.Module()
        model.emb = ParallaxEmbedding(50, 4)
        opt = optim.Adagrad(model.parameters(), lr=0.5)
        eng = SparsePSEngine(comm, classify_params(model), opt)
        eng.prepare_step()
        ids = torch.tensor([2, 3, 4]) if rank == 0 else torch.tensor([7])
        model.emb(ids).sum().backward()
        eng.reduce_and_apply()
        assert eng._counts_static is False
        got = [torch.zeros_like(model.emb.weight) for _ in range(world)]
        torch.distributed.all_gather(got, model.emb.weight.detach())
        assert torch.allclose(got[0], got[1], atol=1e-6)
        eng.close()
    finally:
        os.environ.pop('PARALLAX_PADDED_SPARSE', None)


def test_padded_exchange_world2():
    _spawn(_body_padded_exchange_equivalence, get_free_port())


def test_world8_full_stack(free_port):
    """Scaling-path insurance at the driver's maximum world size: the
    hybrid engine stack (dense buckets + sparse PS exchange + optimizer)
    at world_size=8 on gloo, tiny model — exercises bucket membership,
    padded-count verification and collective ordering at the exact rank
    count of the round-end 8-GPU run."""
    _spawn(_body_parallel_run_hybrid, free_port, world=8)


def _body_densenet_cifar_mpi(rank, world):
    """All-dense cnn graph (densenet40_k12: ~120 small conv/BN params)
    through the full parallel_run MPI path — exercises bucket packing
    over many small tensors and the degenerate no-sparse dispatch."""
    import parallax_amd
    from parallax_amd.config import ParallaxConfig
    from parallax_amd.models.cnn import build_cnn_graph

    def build():
        torch.manual_seed(31 + rank)   # broadcast must fix divergent init
        return build_cnn_graph('densenet40_k12', num_classes=11,
                               batch_size=4, optimizer_name='sgd', lr=0.05,
                               use_bf16=False, channels_last=False,
                               dataset='cifar10')

    sess, num_workers, worker_id, _ = parallax_amd.parallel_run(
        build, None, parallax_config=ParallaxConfig(run_option='MPI'))
    assert num_workers == world and worker_id == rank
    torch.manual_seed(200 + rank)      # different data per rank
    x = torch.randn(4, 3, 32, 32)
    y = torch.randint(0, 11, (4,))
    for _ in range(2):
        out = sess.run((x, y))
        assert torch.isfinite(torch.as_tensor(out['loss']))
    model = sess.graph.model
    # every parameter bitwise-synced across ranks after averaged steps
    for name, p in list(model.named_parameters())[::17]:
        ps = [torch.zeros_like(p) for _ in range(world)]
        torch.distributed.all_gather(ps, p.detach())
        assert torch.allclose(ps[0], ps[1], atol=1e-5), name
    sess.close()


def test_densenet_cifar_mpi_world2():
    _spawn(_body_densenet_cifar_mpi, get_free_port(), world=2, timeout=300)


def _body_local_aggregation_off(rank, world):
    """local_aggregation=False (reference graph_transform_lib:1544-1549
    bypass): raw slices go to the exchange un-deduplicated; the final
    table must match the aggregated path exactly."""
    import torch.nn as nn
    from parallax_amd import optim
    from parallax_amd.parallel.comm import CommContext
    from parallax_amd.parallel.sparse_engine import SparsePSEngine
    from parallax_amd.ops.embedding import ParallaxEmbedding
    from parallax_amd.tagging import classify_params

    comm = CommContext()
    tables = {}
    for local_agg in (True, False):
        torch.manual_seed(17)
        model = nn.Module()
        model.emb = ParallaxEmbedding(30, 4)
        opt = optim.Adagrad(model.parameters(), lr=0.5)
        eng = SparsePSEngine(comm, classify_params(model), opt,
                             local_aggregation=local_agg)
        eng.prepare_step()
        ids = torch.tensor([2 + rank, 9, 9, 2 + rank])  # dups + overlap
        model.emb(ids).sum().backward()
        eng.reduce_and_apply()
        tables[local_agg] = model.emb.weight.detach().clone()
        # ranks agree within each mode
        ws = [torch.zeros_like(tables[local_agg]) for _ in range(world)]
        torch.distributed.all_gather(ws, tables[local_agg])
        assert torch.allclose(ws[0], ws[1], atol=1e-6)
        eng.close()
    assert torch.allclose(tables[True], tables[False], atol=1e-6), \
        (tables[True] - tables[False]).abs().max()


def test_local_aggregation_off_world2():
    _spawn(_body_local_aggregation_off, get_free_port(), world=2,
           timeout=300)
