"""parallel_run — the public entry point (master spawner + worker setup).

MI355X-native equivalent of the reference's common/runner.py:139-193 plus
the per-mode launch/run halves (mpi/ps/hybrid runner.py).  The launch plane
is a LOCAL per-GPU process spawner (one rank per GPU over RCCL/xGMI)
instead of ssh+mpirun: the master process re-execs the user's driver script
once per GPU with the env-var role protocol (consts.py), each worker
re-enters parallel_run, takes the worker branch and gets back a
ParallaxSession.  Running under an external launcher (torchrun/the bench
driver) is detected via WORLD_SIZE/RANK and skips the spawn.

The partition-search outer loop (reference runner.py:73-128 +
partitions.py) restarts the whole job with PARALLAX_PARTITIONS=p, collects
the step-50..100 mean iteration time from every worker over a local
BaseManager queue, and walks p via PartitionStatCollector.
"""

import os
import signal
import subprocess
import sys
import time
from multiprocessing.managers import BaseManager

import torch

from parallax_amd import consts
from parallax_amd.config import ParallaxConfig
from parallax_amd.lib import (parallax_log, parse_resource_info,
                              serialize_resource_info,
                              deserialize_resource_info, num_total_workers,
                              get_free_port, redirect_stdout_stderr)
from parallax_amd.parallel.comm import CommContext
from parallax_amd.parallel.hybrid import HybridEngine
from parallax_amd.partitions import PartitionStatCollector
from parallax_amd.session import ParallaxSession, TrainGraph
from parallax_amd.tagging import classify_params
from parallax_amd.consts import (PARALLAX_MIN_PARTITIONS,
                                 PARALLAX_PARTITIONS, PARALLAX_SEARCH)


class _StatQueueManager(BaseManager):
    pass


def _make_master_queue(port):
    import queue as _q
    q = _q.Queue()
    _StatQueueManager.register('queue', callable=lambda: q)
    m = _StatQueueManager(address=('127.0.0.1', port),
                          authkey=b'parallax_auth')
    m.start()
    return m


def _connect_worker_queue():
    addr = os.environ.get('PARALLAX_STAT_ADDR')
    if not addr:
        return None
    host, port = addr.rsplit(':', 1)
    _StatQueueManager.register('queue')
    m = _StatQueueManager(address=(host, int(port)),
                          authkey=b'parallax_auth')
    try:
        m.connect()
        return m.queue()
    except Exception as e:
        parallax_log.warning('stat queue connect failed: %r', e)
        return None


def _is_worker():
    return (consts.PARALLAX_RUN_OPTION in os.environ
            and os.environ[consts.PARALLAX_RUN_OPTION]
            != consts.PARALLAX_RUN_MASTER) or 'WORLD_SIZE' in os.environ


# ---------------------------------------------------------------------------
# master half
# ---------------------------------------------------------------------------

def _spawn_workers(resource_info, config, p_to_test, stat_addr):
    """Fork one worker process per local GPU (or one CPU worker when the
    host has none).  Reference analog: mpirun/ssh spawn
    (mpi/runner.py:87-110, ps/runner.py:163-193)."""
    workers = resource_info['worker']
    if len(workers) > 1:
        parallax_log.warning(
            'multiple hosts in resource_info; this MI355X launcher runs the '
            'FIRST (local) host only — use an external launcher for '
            'multi-node')
    w = workers[0]
    n = max(1, len(w['gpus']))
    master_port = get_free_port()
    procs = []
    run_option = {'MPI': consts.PARALLAX_RUN_MPI,
                  'PS': consts.PARALLAX_RUN_PS,
                  'HYBRID': consts.PARALLAX_RUN_HYBRID}[config.run_option]
    for rank in range(n):
        env = dict(os.environ)
        env.update({
            'RANK': str(rank),
            'LOCAL_RANK': str(rank),
            'WORLD_SIZE': str(n),
            consts.MASTER_ADDR: consts.DEFAULT_MASTER_ADDR,
            consts.MASTER_PORT: str(master_port),
            consts.PARALLAX_RUN_OPTION: run_option,
            consts.PARALLAX_WORKER_ID: str(rank),
            consts.PARALLAX_NUM_WORKERS: str(n),
            consts.PARALLAX_MACHINE_ID: '0',
            consts.PARALLAX_HOSTNAME: w['hostname'],
            consts.PARALLAX_RESOURCE_INFO:
                serialize_resource_info(resource_info),
            consts.PARALLAX_SYNC: '1' if config.sync else '0',
        })
        if stat_addr:
            env['PARALLAX_STAT_ADDR'] = stat_addr
        if p_to_test is not None:
            env[PARALLAX_PARTITIONS] = str(int(p_to_test))
            env[PARALLAX_SEARCH] = '1'
        cmd = [sys.executable] + sys.argv
        proc = subprocess.Popen(cmd, env=env, preexec_fn=os.setsid)
        procs.append(proc)

    def cleanup(signum, frame):
        for p in procs:
            try:
                os.killpg(os.getpgid(p.pid), signal.SIGTERM)
            except (ProcessLookupError, OSError):
                pass
        if signum is not None:  # invoked as a signal handler: fail-stop
            sys.exit(128 + signum)

    # fail-stop process-group cleanup on interrupt (reference cleanup_mpi /
    # cleanup_ps, mpi/runner.py:122-130, ps/runner.py:186-192)
    signal.signal(signal.SIGINT, cleanup)
    signal.signal(signal.SIGTERM, cleanup)
    return procs, cleanup


def _run_master(config):
    resource_info = config.resource_info
    num_workers = num_total_workers(resource_info)

    search_p = False
    p_to_test = None
    stat_mgr = None
    stat_port = None
    if config.search_partitions and PARALLAX_MIN_PARTITIONS in os.environ:
        p_to_test = max(int(os.environ[PARALLAX_MIN_PARTITIONS]),
                        num_workers)
        collector = PartitionStatCollector(p_to_test)
        search_p = True
        stat_port = get_free_port()

    cleanup = None
    try:
        while True:
            stat_addr = None
            if search_p:
                stat_mgr = _make_master_queue(stat_port)
                stat_addr = '127.0.0.1:%d' % stat_port
            procs, cleanup = _spawn_workers(resource_info, config,
                                            p_to_test, stat_addr)
            if not search_p:
                # fail-stop: any worker dying non-zero kills the job
                # (reference master wait()s the first process,
                # runner.py:123-124; polling all is strictly safer)
                rc = 0
                running = list(procs)
                while running:
                    time.sleep(0.5)
                    for p in list(running):
                        r = p.poll()
                        if r is None:
                            continue
                        running.remove(p)
                        if r != 0:
                            parallax_log.error(
                                'worker pid %d exited rc=%d; stopping '
                                'the job', p.pid, r)
                            cleanup(None, None)
                            rc = r
                            running = []
                            break
                sys.exit(rc if rc else 0)
            # collect one exec time per worker, watching for deaths
            times, alive = [], True
            while len(times) < num_workers and alive:
                time.sleep(1)
                q = stat_mgr.queue()
                while q.qsize() > 0:
                    times.append(q.get()[1])
                for p in procs:
                    if p.poll() is not None and p.returncode != 0:
                        alive = False
                        break
            cleanup(None, None)
            for p in procs:
                try:
                    p.wait(timeout=30)
                except subprocess.TimeoutExpired:
                    pass
            stat_mgr.shutdown()
            stat_mgr = None
            mean_t = (sum(times) / len(times)) if times else None
            search_p, p_to_test = collector.update(mean_t, alive=alive)
            parallax_log.info('partition search: next p=%s (continue=%s)',
                              p_to_test, search_p)
            if not search_p:
                # final run with the optimal p
                os.environ[PARALLAX_PARTITIONS] = str(int(p_to_test))
                procs, cleanup = _spawn_workers(resource_info, config,
                                                p_to_test, None)
                procs[0].wait()
                break
    finally:
        if cleanup is not None:
            cleanup(None, None)
        if stat_mgr is not None:
            stat_mgr.shutdown()
    sys.exit(0)


# ---------------------------------------------------------------------------
# worker half
# ---------------------------------------------------------------------------

def _run_worker(build_fn, config):
    worker_id = int(os.environ.get(consts.PARALLAX_WORKER_ID,
                                   os.environ.get('RANK', '0')))
    num_workers = int(os.environ.get(consts.PARALLAX_NUM_WORKERS,
                                     os.environ.get('WORLD_SIZE', '1')))
    if config.redirect_path:
        redirect_stdout_stderr(config.redirect_path, worker_id)

    comm = CommContext(rank=worker_id, world_size=num_workers)
    if comm.device.type == 'cuda':
        # let MIOpen's find phase pick tuned conv solvers per shape
        torch.backends.cudnn.benchmark = True
    # Per-rank RNG streams: data/candidate sampling must differ across
    # ranks (the rank-0 broadcast below unifies the model init).
    seed = int(os.environ.get('PARALLAX_SEED', '1234'))
    torch.manual_seed(seed + worker_id)
    graph = build_fn()
    if not isinstance(graph, TrainGraph):
        model, optimizer, step_fn = graph
        graph = TrainGraph(model, optimizer, step_fn)
    graph.model.to(comm.device)
    _move_optimizer_state(graph.optimizer, comm.device)

    # dense/sparse tagging pass (contract N1)
    infos = classify_params(graph.model)
    # rank-0 weight broadcast at init (contract N5)
    comm.broadcast_parameters(
        list(graph.model.named_parameters()) +
        list(graph.model.named_buffers()))
    engine = HybridEngine(comm, infos, graph.optimizer, config)
    if config.export_graph_path and worker_id == 0:
        _export_graph(config.export_graph_path, graph, infos, engine, config)
    stat_queue = _connect_worker_queue()
    sess = ParallaxSession(graph, engine, comm, config, num_workers,
                           worker_id, stat_queue=stat_queue)
    # reference parallel_run returns
    # (sess, num_workers, worker_id, num_replicas_per_worker); one process
    # per GPU -> exactly one replica per worker.
    return sess, num_workers, worker_id, 1


def _export_graph(path, graph, infos, engine, config):
    """Dump the transformed-training-plan description (the reference's
    export_graph_path wrote the transformed MetaGraphDef,
    mpi/runner.py:164-168): model structure, the dense/sparse tagging
    table, engine mode and the allreduce bucket plan."""
    import json
    os.makedirs(path, exist_ok=True)
    with open(os.path.join(path, 'model.txt'), 'w') as f:
        f.write(repr(graph.model))
    plan = {
        'mode': engine.mode,
        'run_option': config.run_option,
        'average_sparse': config.average_sparse,
        'params': [{
            'name': gi.name,
            'shape': list(gi.param.shape),
            'dtype': str(gi.param.dtype),
            'grad': 'INDEXED_SLICES' if gi.is_sparse else 'DENSE',
            'partition': list(gi.partition) if gi.partition else None,
        } for gi in infos],
    }
    if engine.dense_engine is not None:
        plan['allreduce_buckets'] = [
            {'numel': b.numel,
             'params': [p.shape[0] if p.dim() else 1 for p in b.params]}
            for b in engine.dense_engine._buckets]
    with open(os.path.join(path, 'comm_plan.json'), 'w') as f:
        json.dump(plan, f, indent=2, default=str)
    parallax_log.info('exported training plan to %s', path)


def _move_optimizer_state(optimizer, device):
    if optimizer is None:
        return
    for state in optimizer.state.values():
        for k, v in state.items():
            if isinstance(v, torch.Tensor):
                state[k] = v.to(device)


# ---------------------------------------------------------------------------
# public API
# ---------------------------------------------------------------------------

def parallel_run(build_fn, resource_info, sync=True, parallax_config=None):
    """Run a single-GPU training definition data-parallel on this node.

    Args (reference runner.py:139-193 signature, graph -> build_fn):
      build_fn:       callable() -> TrainGraph (or (model, optimizer,
                      step_fn) tuple) — the single-GPU 'graph'.
      resource_info:  path to a resource_info file (reference format
                      ``host[:gpu,...]`` per line) or a literal string.
      sync:           synchronous training (async PS is not part of the
                      single-node architecture; sync=False raises).
      parallax_config: ParallaxConfig.

    Master process: spawns one worker per GPU and exits when they finish.
    Worker process: returns (session, num_workers, worker_id,
    num_replicas_per_worker).
    """
    if not sync:
        raise NotImplementedError(
            'async training is not supported by the MI355X engine '
            '(the reference supports it only in PS mode)')
    config = parallax_config or ParallaxConfig()
    config.set_sync(sync)

    if _is_worker():
        if consts.PARALLAX_RESOURCE_INFO in os.environ:
            config.set_resource_info(deserialize_resource_info(
                os.environ[consts.PARALLAX_RESOURCE_INFO]))
        return _run_worker(build_fn, config)

    config.set_resource_info(
        parse_resource_info(resource_info, config.run_option))
    _run_master(config)  # never returns (sys.exit)
