"""RCCL-branch functional tests WITHOUT an 8-GPU node: two ranks share
cuda:0 (functional, not performant), exercising every nccl-only branch
that otherwise runs only under gloo/CPU:

  * all_to_all_v -> all_to_all_single with splits (sharded lookup path)
  * bf16 all_gather_v / all_gather_v_multi payloads
  * all_gather_fixed -> all_gather_into_tensor (padded sparse exchange)
  * dense bucketed all-reduce with comm-stream overlap + bf16 fusion
  * _ShardedLookup end-to-end (replicate_variables=False over RCCL)

If this RCCL build refuses two ranks on one device the tests SKIP with
the refusal recorded (the paths then remain gloo-only until the driver
gets a multi-GPU node).
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

from parallax_amd.lib import get_free_port

pytestmark = pytest.mark.gpu

WORLD = 2
_SKIP_MARKERS = ('Duplicate GPU', 'duplicate GPU', 'invalid usage',
                 'ncclInvalidUsage')


def _child(fn, rank, world, port, err_q):
    try:
        os.environ['RANK'] = str(rank)
        os.environ['LOCAL_RANK'] = '0'          # both ranks -> cuda:0
        os.environ['WORLD_SIZE'] = str(world)
        os.environ['MASTER_ADDR'] = '127.0.0.1'
        os.environ['MASTER_PORT'] = str(port)
        from parallax_amd.parallel.comm import CommContext
        comm = CommContext(backend='nccl')
        # probe: the first collective surfaces a duplicate-device refusal
        t = torch.ones(4, device=comm.device)
        torch.distributed.all_reduce(t)
        torch.cuda.synchronize()
        assert float(t.sum()) == 4.0 * world
        fn(comm, rank, world)
        torch.cuda.synchronize()
    except Exception as e:
        import traceback
        msg = traceback.format_exc()
        if any(m in str(e) for m in _SKIP_MARKERS):
            err_q.put('__RCCL_REFUSED__: %s' % (str(e).splitlines()[0]
                                                if str(e) else e))
        else:
            err_q.put(msg)
        raise SystemExit(1)
    finally:
        import torch.distributed as dist
        if dist.is_initialized():
            dist.destroy_process_group()


def _spawn_gpu(fn, world=WORLD, timeout=240):
    if not torch.cuda.is_available():
        pytest.skip('needs a GPU')
    port = get_free_port()
    ctx = mp.get_context('spawn')
    err_q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_child, args=(fn, r, world, port, err_q))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout)
    errs = []
    while not err_q.empty():
        errs.append(str(err_q.get()))
    for p in procs:
        if p.is_alive():
            p.terminate()
            errs.append('worker timed out')
    if any(e.startswith('__RCCL_REFUSED__') for e in errs):
        pytest.skip('this RCCL refuses 2 ranks on one device: %s'
                    % next(e for e in errs
                           if e.startswith('__RCCL_REFUSED__')))
    assert not errs and all(p.exitcode == 0 for p in procs), \
        '\n'.join(errs) or [p.exitcode for p in procs]


# --- bodies (module-level: picklable under spawn) --------------------------

def _body_collectives(comm, rank, world):
    # all_gather_v: variable per-rank lengths, bf16 payloads
    n = 3 + 2 * rank
    t = torch.full((n, 5), float(rank + 1), dtype=torch.bfloat16,
                   device=comm.device)
    outs = comm.all_gather_v(t)
    assert [o.shape[0] for o in outs] == [3, 5]
    for r, o in enumerate(outs):
        assert o.dtype == torch.bfloat16
        assert torch.all(o == float(r + 1))
    # zero-length contribution from rank 1
    t2 = torch.ones(0 if rank == 1 else 2, 4, device=comm.device)
    outs = comm.all_gather_v(t2)
    assert outs[0].shape[0] == 2 and outs[1].shape[0] == 0
    # all_gather_v_multi: ids + bf16 rows + counts, single counts exchange
    ids = torch.arange(n, device=comm.device)
    rows = torch.randn(n, 4, device=comm.device).bfloat16()
    cnts = torch.ones(n, device=comm.device)
    g_ids, g_rows, g_cnts = comm.all_gather_v_multi([ids, rows, cnts])
    assert [x.shape[0] for x in g_ids] == [3, 5]
    assert g_rows[rank].dtype == torch.bfloat16
    # all_gather_fixed: nccl all_gather_into_tensor branch
    t3 = torch.full((4, 3), float(rank), device=comm.device)
    cat = comm.all_gather_fixed(t3)
    assert cat.shape == (8, 3)
    assert torch.all(cat[:4] == 0.0) and torch.all(cat[4:] == 1.0)
    # all_to_all_v: nccl all_to_all_single with splits (incl. a 0-count)
    send = [torch.full((rank + r, 2), 10.0 * rank + r, device=comm.device)
            for r in range(world)]
    got = comm.all_to_all_v(send)
    for src, o in enumerate(got):
        assert o.shape[0] == src + rank
        assert torch.all(o == 10.0 * src + rank)


def _body_dense_engine_bf16_overlap(comm, rank, world):
    import torch.nn as nn
    from parallax_amd.parallel.dense_engine import DenseAllReduceEngine
    from parallax_amd.tagging import classify_params

    torch.manual_seed(5)
    model = nn.Sequential(nn.Linear(16, 32), nn.ReLU(),
                          nn.Linear(32, 8)).to(comm.device)
    infos = classify_params(model)
    eng = DenseAllReduceEngine(comm, infos, bucket_bytes=512,  # >1 bucket
                               overlap=True, comm_dtype=torch.bfloat16)
    for step in range(2):
        eng.prepare_step()
        model.zero_grad(set_to_none=True)
        torch.manual_seed(100 * step + rank)
        x = torch.randn(8, 16, device=comm.device)
        loss = model(x).pow(2).sum()
        loss.backward()
        eng.reduce_gradients()
        for p in model.parameters():
            got = [torch.zeros_like(p.grad) for _ in range(world)]
            torch.distributed.all_gather(got, p.grad.contiguous())
            assert torch.allclose(got[0], got[1], atol=1e-5)
    eng.close()


def _body_unused_param_nccl(comm, rank, world):
    import torch.nn as nn
    from parallax_amd.parallel.dense_engine import DenseAllReduceEngine
    from parallax_amd.tagging import classify_params

    torch.manual_seed(9)
    model = nn.Module()
    model.a = nn.Linear(8, 8)
    model.b = nn.Linear(8, 8)       # unused on rank 1, step 0
    model.c = nn.Linear(8, 2)
    model.to(comm.device)
    eng = DenseAllReduceEngine(comm, classify_params(model),
                               bucket_bytes=128, overlap=True)
    eng.prepare_step()
    x = torch.randn(4, 8, device=comm.device)
    h = model.a(x)
    if rank == 0:
        h = model.b(h)
    model.c(h).pow(2).sum().backward()
    eng.reduce_gradients()
    for p in model.parameters():
        assert p.grad is not None
        got = [torch.zeros_like(p.grad) for _ in range(world)]
        torch.distributed.all_gather(got, p.grad.contiguous())
        assert torch.allclose(got[0], got[1], atol=1e-6)
    eng.close()


def _body_sharded_lookup(comm, rank, world):
    import torch.nn as nn
    from parallax_amd import optim
    from parallax_amd.parallel.sparse_engine import SparsePSEngine
    from parallax_amd.ops.embedding import PartitionedEmbedding
    from parallax_amd.tagging import classify_params
    from parallax_amd.partitions import FixedSizePartitioner

    torch.manual_seed(21)
    model = nn.Module()
    model.emb = PartitionedEmbedding(64, 8, FixedSizePartitioner(4))
    model.to(comm.device)
    full_before = model.emb.weight.detach().clone()
    opt = optim.Adagrad(model.parameters(), lr=0.4)
    eng = SparsePSEngine(comm, classify_params(model), opt,
                         local_aggregation=True,
                         replicate_variables=False)
    assert model.emb._sharded
    eng.prepare_step()
    ids = torch.tensor([1, 17, 33, 49, 1 + rank],
                       device=comm.device)
    out = model.emb(ids)
    torch.cuda.synchronize()
    # forward fetched the right rows from their owner ranks
    assert torch.allclose(out.detach(), full_before[ids.cpu()].to(
        comm.device), atol=1e-6)
    out.sum().backward()
    eng.reduce_and_apply()
    # tables on both ranks agree after the update
    full_after = model.emb.full_weight()
    got = [torch.zeros_like(full_after) for _ in range(world)]
    torch.distributed.all_gather(got, full_after.contiguous())
    assert torch.allclose(got[0], got[1], atol=1e-5)
    # touched rows moved, untouched unchanged
    assert not torch.allclose(full_after[1], full_before[1])
    assert torch.allclose(full_after[5], full_before[5])
    eng.close()


def _body_padded_exchange_nccl(comm, rank, world):
    import torch.nn as nn
    from parallax_amd import optim
    from parallax_amd.parallel.sparse_engine import SparsePSEngine
    from parallax_amd.ops.embedding import ParallaxEmbedding
    from parallax_amd.tagging import classify_params

    results = {}
    for mode in ('1', '0'):   # padded (default on GPU) vs dynamic
        os.environ['PARALLAX_PADDED_SPARSE'] = mode
        try:
            torch.manual_seed(31)
            model = nn.Module()
            model.emb = ParallaxEmbedding(50, 4)
            model.to(comm.device)
            opt = optim.Adagrad(model.parameters(), lr=0.5)
            eng = SparsePSEngine(comm, classify_params(model), opt,
                                 average_sparse=True)
            eng.prepare_step()
            ids = torch.tensor([1 + rank, 5, 5, 30 + 2 * rank],
                               device=comm.device)
            model.emb(ids).sum().backward()
            eng.reduce_and_apply()
            if mode == '1':
                assert eng._padded and eng._counts_static is True
            results[mode] = model.emb.weight.detach().clone()
            eng.close()
        finally:
            os.environ.pop('PARALLAX_PADDED_SPARSE', None)
    assert torch.allclose(results['1'], results['0'], atol=1e-4), \
        (results['1'] - results['0']).abs().max()


def test_rccl_collectives_2ranks_1gpu():
    _spawn_gpu(_body_collectives)


def test_rccl_dense_engine_2ranks_1gpu():
    _spawn_gpu(_body_dense_engine_bf16_overlap)


def test_rccl_unused_param_2ranks_1gpu():
    _spawn_gpu(_body_unused_param_nccl)


def test_rccl_sharded_lookup_2ranks_1gpu():
    _spawn_gpu(_body_sharded_lookup)


def test_rccl_padded_exchange_2ranks_1gpu():
    _spawn_gpu(_body_padded_exchange_nccl)
