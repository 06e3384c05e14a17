"""Checkpoint / resume — chief-only, name-keyed, partition-shard aware.

Reproduces the reference checkpoint contract (reference
common/lib.py:38-56 build_ckpt_hooks + MonitoredTrainingSession auto-restore):
  * keyed by the USER-VISIBLE parameter names of the single-GPU model
    (replica transforms never rename variables), including partitioned
    shards (``emb.parts.0`` etc. <-> the reference's ``emb/part_0``);
  * saved by the chief (worker 0) only, on a step and/or seconds cadence;
  * restore is implicit at session start when a checkpoint exists;
  * max_to_keep is effectively unlimited (reference max_to_keep=1000000).

Format: one ``model.ckpt-<global_step>.pt`` torch.save archive per save
({'model', 'optimizer', 'global_step'}) plus a ``checkpoint`` pointer file
naming the latest (the TF checkpoint-file convention).
"""

import os
import time

import torch

from parallax_amd.lib import parallax_log


class CheckpointManager(object):
    def __init__(self, ckpt_config, model, optimizer, is_chief):
        self.cfg = ckpt_config
        self.model = model
        self.optimizer = optimizer
        self.is_chief = is_chief
        self.enabled = (ckpt_config is not None
                        and ckpt_config.ckpt_dir is not None
                        and (ckpt_config.save_ckpt_steps is not None
                             or ckpt_config.save_ckpt_secs is not None))
        self._last_save_time = time.time()
        if self.enabled and self.is_chief:
            os.makedirs(self.cfg.ckpt_dir, exist_ok=True)

    def _sharded(self):
        """True when any table uses shard-only storage
        (replicate_variables=False): state_dict() is then a COLLECTIVE
        and every rank holds rank-local optimizer state."""
        return any(getattr(m, '_sharded', False)
                   for m in self.model.modules())

    def _rank_world(self):
        import torch.distributed as dist
        if dist.is_initialized():
            return dist.get_rank(), dist.get_world_size()
        return 0, 1

    # -- save ----------------------------------------------------------------
    def maybe_save(self, global_step):
        if not self.enabled:
            return False
        sharded = self._sharded()
        if not (self.is_chief or sharded):
            return False
        due = False
        if (self.cfg.save_ckpt_steps
                and global_step % self.cfg.save_ckpt_steps == 0
                and global_step > 0):
            due = True
        if self.cfg.save_ckpt_secs:
            secs_due = (time.time() - self._last_save_time
                        >= self.cfg.save_ckpt_secs)
            if sharded:
                # wall-clock cadence diverges across ranks and the save
                # is collective in sharded mode: EVERY rank must enter
                # the broadcast every step (a rank-local condition would
                # desynchronize the collective), the chief decides
                due = self._bcast_flag(due or secs_due)
            else:
                due = due or secs_due
        if due:
            self.save(global_step)
        return due

    def _bcast_flag(self, flag):
        import torch.distributed as dist
        # NCCL/RCCL (all GPU runs) broadcasts device tensors only; gloo
        # wants CPU.  A CPU flag under nccl raises at dist.broadcast.
        dev = (torch.device('cuda', torch.cuda.current_device())
               if dist.get_backend() == 'nccl' else torch.device('cpu'))
        t = torch.tensor([1 if (flag and self.is_chief) else 0],
                         dtype=torch.long, device=dev)
        dist.broadcast(t, src=0)
        return bool(t.item())

    def save(self, global_step):
        sharded = self._sharded()
        rank, world = self._rank_world()
        path = os.path.join(self.cfg.ckpt_dir,
                            'model.ckpt-%d.pt' % global_step)
        # model state_dict: collective in sharded mode (gathers shards);
        # every rank builds it, only the chief writes it
        model_sd = {k: v.detach().cpu()
                    for k, v in self.model.state_dict().items()}
        if sharded and self.optimizer is not None:
            # per-row optimizer state is rank-local: each rank writes its
            # own sidecar; restore requires the same world size
            side = os.path.join(
                self.cfg.ckpt_dir, 'model.ckpt-%d.optshard%d-of-%d.pt'
                % (global_step, rank, world))
            torch.save(self.optimizer.state_dict(), side + '.tmp')
            os.replace(side + '.tmp', side)
        if not self.is_chief:
            self._last_save_time = time.time()
            return
        payload = {
            'model': model_sd,
            'optimizer': (None if sharded else
                          self.optimizer.state_dict()
                          if self.optimizer is not None else None),
            'sharded_world': world if sharded else None,
            'global_step': global_step,
        }
        tmp = path + '.tmp'
        torch.save(payload, tmp)
        os.replace(tmp, path)
        with open(os.path.join(self.cfg.ckpt_dir, 'checkpoint'), 'w') as f:
            f.write('model_checkpoint_path: "%s"\n'
                    % os.path.basename(path))
        self._last_save_time = time.time()
        parallax_log.info('saved checkpoint %s', path)

    # -- restore -------------------------------------------------------------
    def latest_path(self):
        if not self.enabled:
            return None
        pointer = os.path.join(self.cfg.ckpt_dir, 'checkpoint')
        if not os.path.isfile(pointer):
            return None
        with open(pointer) as f:
            for line in f:
                if line.startswith('model_checkpoint_path:'):
                    name = line.split('"')[1]
                    return os.path.join(self.cfg.ckpt_dir, name)
        return None

    def maybe_restore(self, map_location='cpu'):
        """Restore latest checkpoint if one exists (all ranks restore the
        same file — reference MonitoredTrainingSession restores on chief and
        broadcasts; loading the same file everywhere is equivalent and has
        no broadcast step).  Returns restored global_step or 0."""
        path = self.latest_path()
        if path is None:
            return 0
        payload = torch.load(path, map_location=map_location,
                             weights_only=False)
        self.model.load_state_dict(payload['model'])
        if self.optimizer is not None and payload.get('optimizer'):
            self.optimizer.load_state_dict(payload['optimizer'])
        elif self.optimizer is not None and payload.get('sharded_world'):
            rank, world = self._rank_world()
            side = path.replace('.pt', '.optshard%d-of-%d.pt'
                                % (rank, world))
            if payload['sharded_world'] == world and os.path.isfile(side):
                self.optimizer.load_state_dict(
                    torch.load(side, map_location=map_location,
                               weights_only=False))
            else:
                parallax_log.warning(
                    'sharded checkpoint written at world=%s, running at '
                    'world=%d: per-row optimizer state NOT restored '
                    '(weights restored; accumulators start fresh)',
                    payload['sharded_world'], world)
        parallax_log.info('restored checkpoint %s (global_step=%d)',
                          path, payload['global_step'])
        return int(payload['global_step'])
