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

    # -- the step ------------------------------------------------------------
    def _run_step(self, batch):
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

    def close(self):
        self.engine.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False
