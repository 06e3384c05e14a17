"""Input-pipeline prefetch — the StagingArea/QueueRunner equivalent.

Contract N9: the reference double-buffers host->device
transfer through TF StagingArea ops (benchmark_cnn.py:690-703) so the GPU
never waits on input.  MI355X-native equivalent: pinned host staging +
``hipMemcpyAsync`` on a dedicated copy stream, one batch ahead
(double-buffered), with an event handoff to the compute stream.
"""

import torch

from parallax_amd.shard import shard as _shard


class PrefetchLoader(object):
    """Wrap any batch iterable; yields device batches copied one step
    ahead on a side HIP stream.  Tuples/lists/dicts of tensors supported.
    On CPU devices it is a passthrough."""

    def __init__(self, loader, device, pin_memory=True):
        self.loader = loader
        self.device = torch.device(device)
        self.pin = pin_memory and self.device.type == 'cuda'
        self.stream = (torch.cuda.Stream(device=self.device)
                       if self.device.type == 'cuda' else None)

    def _to_device(self, batch):
        def mv(t):
            if not isinstance(t, torch.Tensor):
                return t
            if self.pin and t.device.type == 'cpu' and not t.is_pinned():
                t = t.pin_memory()
            return t.to(self.device, non_blocking=True)

        if isinstance(batch, (tuple, list)):
            return type(batch)(mv(b) for b in batch)
        if isinstance(batch, dict):
            return {k: mv(v) for k, v in batch.items()}
        return mv(batch)

    def _record(self, batch, stream):
        if isinstance(batch, (tuple, list)):
            for b in batch:
                self._record(b, stream)
        elif isinstance(batch, dict):
            for b in batch.values():
                self._record(b, stream)
        elif isinstance(batch, torch.Tensor) and batch.device.type == 'cuda':
            batch.record_stream(stream)

    def __iter__(self):
        if self.stream is None:
            for batch in self.loader:
                yield self._to_device(batch)
            return
        it = iter(self.loader)
        next_batch = None
        next_event = None
        try:
            host = next(it)
        except StopIteration:
            return
        with torch.cuda.stream(self.stream):
            next_batch = self._to_device(host)
            next_event = torch.cuda.Event()
            next_event.record(self.stream)
        while True:
            batch, ev = next_batch, next_event
            try:
                host = next(it)
            except StopIteration:
                host = None
            if host is not None:
                with torch.cuda.stream(self.stream):
                    next_batch = self._to_device(host)
                    next_event = torch.cuda.Event()
                    next_event.record(self.stream)
            # compute stream waits for the copy of the CURRENT batch only
            ev.wait(torch.cuda.current_stream())
            # Tie each yielded tensor's allocation to the consumer stream:
            # without record_stream the caching allocator may hand the
            # memory back to the NEXT side-stream prefetch copy while
            # compute-stream kernels still read this batch (the classic
            # apex/torchvision prefetcher hazard).
            self._record(batch, torch.cuda.current_stream())
            yield batch
            if host is None:
                return


def sharded_prefetch_loader(dataset_iterable, device, num_shards=None,
                            shard_id=None):
    """shard() + PrefetchLoader composed — the per-worker input pipeline
    (reference shard.shard + StagingArea)."""
    return PrefetchLoader(_shard(dataset_iterable, num_shards, shard_id),
                          device)
