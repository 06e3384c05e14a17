"""ParallaxSession — the per-worker training handle returned by parallel_run.

MI355X-native equivalent of the reference's monkey-patched Session.run
(common/session_context.py:35-92,235-255): the user's single-GPU step runs
unchanged; the session adds around it
  * the engine protocol (prepare -> forward -> backward -> comm -> update);
  * a global step counter;
  * per-step profiling on the configured worker/steps (ProfileConfig ->
    torch.profiler trace dumps, the rocprof-era analog of the reference's
    RunMetadata FULL_TRACE dumps, session_context.py:74-92);
  * exec-time stat collection over steps 50..100 reported to the master's
    queue for partition search (session_context.py:28-29,54-71);
  * chief-only checkpointing on the reference cadence.
"""

import os
import time

import torch

from parallax_amd import consts
from parallax_amd.checkpoint import CheckpointManager
from parallax_amd.lib import parallax_log


class TrainGraph(object):
    """What build_fn returns: the single-GPU 'graph'.

    model:      nn.Module (on the worker's device after parallel_run).
    optimizer:  a parallax_amd.optim optimizer over model.parameters().
    step_fn:    callable(model, batch) -> (loss_tensor, fetches_dict).
    """

    def __init__(self, model, optimizer, step_fn, lr_scheduler=None):
        self.model = model
        self.optimizer = optimizer
        self.step_fn = step_fn
        self.lr_scheduler = lr_scheduler


class ParallaxSession(object):
    def __init__(self, graph, engine, comm, config, num_workers, worker_id,
                 stat_queue=None):
        self.graph = graph
        self.engine = engine
        self.comm = comm
        self.config = config
        self.num_workers = num_workers
        self.worker_id = worker_id
        self.global_step = 0
        self._stat_queue = stat_queue
        self._stat_sent = False
        self._stat_start_time = None
        self.ckpt = CheckpointManager(config.ckpt_config, graph.model,
                                      graph.optimizer,
                                      is_chief=(worker_id == 0))
        self.global_step = self.ckpt.maybe_restore(
            map_location=str(comm.device))
        pc = config.profile_config
        self._profile_dir = None
        if pc.profile_dir is not None and (
                pc.profile_worker is None or pc.profile_worker == worker_id):
            self._profile_dir = os.path.join(
                pc.profile_dir, 'worker:%d' % worker_id)
            os.makedirs(self._profile_dir, exist_ok=True)

    # -- profiling (reference session_context.py:94-147) ---------------------
    def _is_profile_step(self):
        pc = self.config.profile_config
        if self._profile_dir is None:
            return False
        s = self.global_step
        if pc.profile_steps is not None and s in pc.profile_steps:
            return True
        if (pc.profile_range is not None
                and pc.profile_range[0] <= s < pc.profile_range[1]):
            return True
        return False

    def _run_profiled(self, batch):
        from torch.profiler import profile, ProfilerActivity
        acts = [ProfilerActivity.CPU]
        if self.comm.device.type == 'cuda':
            acts.append(ProfilerActivity.CUDA)
        with profile(activities=acts) as prof:
            out = self._run_step(batch)
        trace = os.path.join(self._profile_dir,
                             'trace_%d.json' % self.global_step)
        try:
            prof.export_chrome_trace(trace)
        except Exception as e:  # pragma: no cover
            parallax_log.warning('profile export failed: %r', e)
        return out

    # -- hipGraph capture of the fwd+bwd hot loop ---------------------------
    # The LM1B step is launch-bound (hundreds of small kernels across 20
    # unrolled LSTM steps); capturing forward+backward in ONE hipGraph and
    # replaying it removes the per-kernel host gaps (guide: HIP streams and
    # graphs instead of a tracing compiler).  Outside the graph remain the
    # dynamic-shape parts: the sparse dedup/exchange, collectives, and the
    # optimizer step.
    def enable_cuda_graph(self, example_batch, warmup_iters=3):
        """Capture fwd+bwd (+recurrent-state feedback) as a hipGraph over
        static input buffers.  Requires static shapes and a CUDA device.
        ``example_batch``: (tensors..., state_or_None) matching step_fn."""
        assert self.comm.device.type == 'cuda', 'cuda graphs need a GPU'
        g = self.graph
        if any(getattr(m, '_sharded', False) for m in g.model.modules()):
            raise RuntimeError(
                'hipGraph capture is unavailable with shard-only tables '
                '(replicate_variables=False): the lookup runs collectives '
                'with data-dependent message sizes inside forward')
        if self.engine.dense_engine:
            self.engine.dense_engine.set_overlap(False)

        # static input buffers (state tensors feed back inside the graph)
        def _clone(t):
            return t.clone() if isinstance(t, torch.Tensor) else t
        self._static_batch = tuple(_clone(b) for b in example_batch)

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                self.engine.prepare_step()
                g.optimizer.zero_grad(set_to_none=True)
                loss, fetches = g.step_fn(g.model, self._static_batch)
                loss.backward()
        torch.cuda.current_stream().wait_stream(side)

        # materialize dense grads so the captured zero_()s hit stable memory
        dense_params = [p for p in g.model.parameters()
                        if p.grad is not None
                        and p.grad.layout == torch.strided]
        self.engine.prepare_step()  # clear sparse capture slots pre-capture
        self._graph_obj = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph_obj):
            self._zero_dense_grads(dense_params)
            loss, fetches = g.step_fn(g.model, self._static_batch)
            loss.backward()
            state = fetches.get('state') if fetches else None
            if state is not None and isinstance(
                    self._static_batch[-1], (tuple, list)):
                for dst, src in zip(self._static_batch[-1], state):
                    dst.copy_(src)
        self._static_loss = loss
        self._static_fetches = fetches
        self._opt_graph = None
        self._opt_graph_state = 'pending'
        self._opt_graph_steps = 0
        parallax_log.info('hipGraph capture enabled (%d dense grads, '
                          '%d sparse slots)', len(dense_params),
                          len(self.engine.sparse_engine.slots)
                          if self.engine.sparse_engine else 0)

    @staticmethod
    def _zero_dense_grads(dense_params):
        """Zero the static grad buffers at the top of the captured step:
        ONE fused launch per <=32 same-dtype tensors instead of a
        per-tensor zero_() storm."""
        from parallax_amd import ops as _ops
        if not _ops.native_available():
            for p in dense_params:
                p.grad.zero_()
            return
        by_dtype = {}
        for p in dense_params:
            by_dtype.setdefault(p.grad.dtype, []).append(p.grad.detach())
        for grads in by_dtype.values():
            _ops.ext().multi_tensor_zero(grads)

    def _maybe_capture_opt_graph(self):
        """Capture the (shape-static) optimizer step — fused dense update,
        clip, EMA — as a SECOND hipGraph once its state tensors exist
        (after a few eager steps).  The gradient buffers are stable under
        the main graph, and at world>1 the unpack kernel writes the
        reduced grads into those same buffers before this replays."""
        if self._opt_graph_state == 'disabled':
            return False
        if self._opt_graph_state == 'ready':
            return True
        import os
        if os.environ.get('PARALLAX_OPT_GRAPH', '1') == '0':  # debug/A-B
            self._opt_graph_state = 'disabled'
            return False
        # (Adam is capturable too: its bias-correction step count lives in
        # a 0-dim device tensor the kernel dereferences — optim.Adam)
        self._opt_graph_steps += 1
        if self._opt_graph_steps < 3:
            return False
        # the sync-free padded sparse tail (sort + fixed-shape dedup +
        # masked row update, no collectives at world 1) is captured INTO
        # the optimizer graph: sparse dedup/apply then costs one replay
        # instead of ~15 eager launches per table per step
        sp = self.engine.sparse_engine
        self._sparse_in_opt_graph = self._opt_graph_owns_sparse()
        try:
            # the warm call below EXECUTES this step's real update; the
            # capture that follows only records — so return False this
            # step (the update already happened) and replay from the next
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                if self._sparse_in_opt_graph:
                    sp.reduce_and_apply()
                self.graph.optimizer.step()
            torch.cuda.current_stream().wait_stream(side)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                if self._sparse_in_opt_graph:
                    sp.reduce_and_apply()
                self.graph.optimizer.step()
            self._opt_graph = g
            self._opt_graph_state = 'just_captured'
            parallax_log.info('optimizer-step hipGraph captured%s',
                              ' (sparse tail included)'
                              if self._sparse_in_opt_graph else '')
        except Exception as e:
            parallax_log.warning('optimizer graph capture failed (%r); '
                                 'staying eager', e)
            self._opt_graph_state = 'disabled'
            self._sparse_in_opt_graph = False
        return False

    def _opt_graph_owns_sparse(self):
        """Will the optimizer graph (its warm/capture call this step, or
        its replay) run the sparse tail — so the eager call must skip it?"""
        if self._opt_graph_state == 'disabled':
            return False
        sp = self.engine.sparse_engine
        if sp is None or not sp.capture_safe() \
                or os.environ.get('PARALLAX_SPARSE_GRAPH', '1') == '0':
            return False
        if self._opt_graph_state in ('ready', 'just_captured'):
            return getattr(self, '_sparse_in_opt_graph', False)
        return self._opt_graph_steps + 1 >= 3  # this step captures

    def _run_step_graphed(self, batch):
        # copy dynamic inputs into the static buffers (state feeds back
        # inside the graph; a caller-passed state is ignored after capture)
        for dst, src in zip(self._static_batch, batch):
            if isinstance(dst, torch.Tensor) and isinstance(
                    src, torch.Tensor) and dst.data_ptr() != src.data_ptr():
                dst.copy_(src, non_blocking=True)
        self._graph_obj.replay()
        eng = self.engine
        if eng.dense_engine:
            eng.dense_engine.launch_async()
        if eng.sparse_engine and not self._opt_graph_owns_sparse():
            eng.sparse_engine.reduce_and_apply()
        if eng.dense_engine:
            eng.dense_engine.wait_and_unpack()
        if self._maybe_capture_opt_graph():
            self._opt_graph.replay()
        else:
            # eager optimizer (the capture step's warm call IS this
            # step's update — _maybe_capture_opt_graph returns False then)
            if self._opt_graph_state != 'just_captured':
                self.graph.optimizer.step()
            else:
                self._opt_graph_state = 'ready'
        if self.graph.lr_scheduler is not None:
            self.graph.lr_scheduler.step()
        return self._static_loss, self._static_fetches

    # -- the step ------------------------------------------------------------
    def _run_step(self, batch):
        if getattr(self, '_graph_obj', None) is not None:
            return self._run_step_graphed(batch)
        g = self.graph
        self.engine.prepare_step()
        g.optimizer.zero_grad(set_to_none=True)
        loss, fetches = g.step_fn(g.model, batch)
        loss.backward()
        self.engine.finish_step()
        if g.lr_scheduler is not None:
            g.lr_scheduler.step()
        return loss, fetches

    def run(self, batch=None):
        """Execute one training step; returns a dict with at least
        'loss' and 'global_step' (reference fetch ergonomics)."""
        if self._is_profile_step():
            loss, fetches = self._run_profiled(batch)
        else:
            loss, fetches = self._run_step(batch)
        self.global_step += 1
        self._collect_stats()
        self.ckpt.maybe_save(self.global_step)
        out = {'loss': loss.detach(), 'global_step': self.global_step}
        if fetches:
            out.update(fetches)
        return out

    # -- partition-search stat window
    #    (reference session_context.py:28-29,54-71) --------------------------
    def _collect_stats(self):
        if self._stat_queue is None or self._stat_sent:
            return
        s = self.global_step
        if s == consts.COLLECT_STAT_START_STEP:
            if self.comm.device.type == 'cuda':
                torch.cuda.synchronize()
            self._stat_start_time = time.time()
        elif s == consts.COLLECT_STAT_END_STEP \
                and self._stat_start_time is not None:
            if self.comm.device.type == 'cuda':
                torch.cuda.synchronize()
            n = consts.COLLECT_STAT_END_STEP - consts.COLLECT_STAT_START_STEP
            exec_time = (time.time() - self._stat_start_time) / n
            try:
                self._stat_queue.put((self.worker_id, exec_time))
            except Exception as e:  # pragma: no cover
                parallax_log.warning('stat report failed: %r', e)
            self._stat_sent = True

    def params_stat(self):
        """Gradient/parameter byte counters (reference --params_stat,
        benchmark_cnn.py:785-796,825-827)."""
        from parallax_amd.tagging import classify_params, split_dense_sparse
        infos = classify_params(self.graph.model)
        sparse, dense = split_dense_sparse(infos)
        num_el = sum(gi.param.numel() for gi in infos)
        grad_bytes = sum(gi.param.numel() * gi.param.element_size()
                         for gi in dense)
        return {
            'num_tensors': len(infos),
            'num_dense': len(dense),
            'num_sparse': len(sparse),
            'num_elements': num_el,
            'param_bytes': sum(gi.param.numel() * gi.param.element_size()
                               for gi in infos),
            'grad_bytes': grad_bytes,
        }

    def close(self):
        self.engine.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False
