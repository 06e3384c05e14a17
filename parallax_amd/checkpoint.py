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


def latest_checkpoint(ckpt_dir):
    """Path of the newest checkpoint under ckpt_dir, or None
    (tf.train.latest_checkpoint equivalent: reads the 'checkpoint'
    pointer file the chief writes)."""
    pointer = os.path.join(ckpt_dir, 'checkpoint')
    if not os.path.isfile(pointer):
        return None
    with open(pointer) as f:
        for line in f:
            if line.startswith('model_checkpoint_path:'):
                return os.path.join(ckpt_dir, line.split('"')[1])
    return None


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

    def _opt_param_index(self):
        """param -> flat index in the optimizer state_dict numbering."""
        params = [p for g in self.optimizer.param_groups
                  for p in g['params']]
        return {id(p): i for i, p in enumerate(params)}

    def _sharded_modules(self):
        return [m for m in self.model.modules()
                if getattr(m, '_sharded', False)]

    def _shard_layout(self):
        """Manifest: optimizer-state index -> this rank's shard layout
        (partition offsets/sizes are global; 'owned' is rank-local)."""
        idx_of = self._opt_param_index()
        layout = {}
        for m in self._sharded_modules():
            idx = idx_of.get(id(m.weight))
            if idx is None:
                continue
            layout[idx] = {'offsets': list(m.offsets),
                           'sizes': list(m.sizes),
                           'owned': list(m._owned),
                           'rows': m.num_embeddings}
        return layout

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
            # own sidecar WITH a shard-layout manifest so a restore at a
            # DIFFERENT world size can reassemble the full per-row state
            # and re-slice it for the new placement
            side = os.path.join(
                self.cfg.ckpt_dir, 'model.ckpt-%d.optshard%d-of-%d.pt'
                % (global_step, rank, world))
            torch.save({'opt': self.optimizer.state_dict(),
                        'layout': self._shard_layout()}, side + '.tmp')
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
        return latest_checkpoint(self.cfg.ckpt_dir)

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
            old_world = payload['sharded_world']
            side = path.replace('.pt', '.optshard%d-of-%d.pt'
                                % (rank, world))
            if old_world == world and os.path.isfile(side):
                sd = torch.load(side, map_location=map_location,
                                weights_only=False)
                if isinstance(sd, dict) and 'opt' in sd:
                    sd = sd['opt']
                self.optimizer.load_state_dict(sd)
            elif not self._reshard_restore(path, old_world, map_location):
                parallax_log.warning(
                    'sharded checkpoint written at world=%s, running at '
                    'world=%d: per-row optimizer state NOT restored '
                    '(weights restored; accumulators start fresh)',
                    old_world, world)
        parallax_log.info('restored checkpoint %s (global_step=%d)',
                          path, payload['global_step'])
        return int(payload['global_step'])

    def _reshard_restore(self, path, old_world, map_location):
        """Rebuild this rank's optimizer state from a sharded checkpoint
        written at a DIFFERENT world size: assemble the full per-row
        state from every old rank's sidecar (each carries a layout
        manifest), then slice it for the CURRENT placement.  Dense /
        replicated state is identical across ranks, so old rank 0's copy
        is taken verbatim.  Returns True on success."""
        import copy
        sides = [path.replace('.pt', '.optshard%d-of-%d.pt' % (r, old_world))
                 for r in range(old_world)]
        if not all(os.path.isfile(s) for s in sides):
            return False
        payloads = [torch.load(s, map_location=map_location,
                               weights_only=False) for s in sides]
        if not all(isinstance(p, dict) and 'opt' in p and 'layout' in p
                   for p in payloads):
            return False  # pre-manifest sidecar format
        base = copy.deepcopy(payloads[0]['opt'])
        idx_of = self._opt_param_index()
        mod_by_idx = {idx_of[id(m.weight)]: m
                      for m in self._sharded_modules()
                      if id(m.weight) in idx_of}
        for idx, meta0 in payloads[0]['layout'].items():
            offsets, sizes = meta0['offsets'], meta0['sizes']
            nrows = meta0['rows']
            # assemble the full [V, ...] tensor per row-shaped state key
            full = {}
            for pl in payloads:
                st_r = pl['opt'].get('state', {}).get(idx)
                if st_r is None:
                    continue
                owned_r = pl['layout'][idx]['owned']
                local_rows = sum(sizes[k] for k in owned_r)
                local_off = 0
                for k in owned_r:
                    off, size = offsets[k], sizes[k]
                    for key, t in st_r.items():
                        if (torch.is_tensor(t) and t.dim() >= 1
                                and t.shape[0] == local_rows):
                            dst = full.get(key)
                            if dst is None:
                                dst = torch.empty(
                                    (nrows,) + tuple(t.shape[1:]),
                                    dtype=t.dtype)
                                full[key] = dst
                            dst[off:off + size] = \
                                t[local_off:local_off + size]
                    local_off += size
            mod = mod_by_idx.get(idx)
            st = base.setdefault('state', {}).setdefault(idx, {})
            for key, t in full.items():
                if mod is not None and mod._sharded:
                    pieces = [t[offsets[k]:offsets[k] + sizes[k]]
                              for k in mod._owned]
                    st[key] = (torch.cat(pieces) if pieces else
                               t.new_zeros((0,) + tuple(t.shape[1:])))
                else:  # restoring into a replicated/world-1 session
                    st[key] = t
        self.optimizer.load_state_dict(base)
        parallax_log.info(
            'resharded optimizer sidecars: world %d -> %d',
            old_world, self._rank_world()[1])
        return True
