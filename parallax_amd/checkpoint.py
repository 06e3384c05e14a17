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

    # -- save ----------------------------------------------------------------
    def maybe_save(self, global_step):
        if not (self.enabled and self.is_chief):
            return False
        due = False
        if (self.cfg.save_ckpt_steps
                and global_step % self.cfg.save_ckpt_steps == 0
                and global_step > 0):
            due = True
        if (self.cfg.save_ckpt_secs
                and time.time() - self._last_save_time
                >= self.cfg.save_ckpt_secs):
            due = True
        if due:
            self.save(global_step)
        return due

    def save(self, global_step):
        path = os.path.join(self.cfg.ckpt_dir,
                            'model.ckpt-%d.pt' % global_step)
        payload = {
            'model': {k: v.detach().cpu()
                      for k, v in self.model.state_dict().items()},
            'optimizer': self.optimizer.state_dict()
            if self.optimizer is not None else None,
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
        parallax_log.info('restored checkpoint %s (global_step=%d)',
                          path, payload['global_step'])
        return int(payload['global_step'])
