"""Collective communication context — RCCL over xGMI (gloo on CPU).

MI355X-native replacement for the reference's three communication planes
(SURVEY §5.8): Horovod MPI+NCCL (dense), TF gRPC PS (sparse), ssh/TCP
control.  On one MI355X node all three collapse onto torch.distributed
(backend "nccl" == RCCL on ROCm) over the 7 xGMI links per GPU; CPU
plumbing tests use gloo (BASELINE config 1).

Contract N6 (deterministic collective ordering) is satisfied by issuing all
collectives from Python in registration order on a dedicated comm stream;
contract N4 (variable-length sparse allgather) by ``all_gather_v``
(counts exchange + padded allgather); the PS "transport" (N7) by
``all_to_all_v``.
"""

import datetime
import os

import torch
import torch.distributed as dist

from parallax_amd import consts


class CommContext(object):
    def __init__(self, rank=None, world_size=None, backend=None,
                 device=None, timeout_sec=300):
        if not dist.is_initialized():
            rank = int(os.environ.get('RANK', 0 if rank is None else rank))
            world_size = int(os.environ.get(
                'WORLD_SIZE', 1 if world_size is None else world_size))
            os.environ.setdefault(consts.MASTER_ADDR,
                                  consts.DEFAULT_MASTER_ADDR)
            os.environ.setdefault(consts.MASTER_PORT, '29400')
            if backend is None:
                backend = 'nccl' if torch.cuda.is_available() else 'gloo'
            dist.init_process_group(
                backend=backend, rank=rank, world_size=world_size,
                timeout=datetime.timedelta(seconds=timeout_sec))
        self.rank = dist.get_rank()
        self.world_size = dist.get_world_size()
        self.backend = dist.get_backend()
        if device is None:
            if torch.cuda.is_available():
                local = int(os.environ.get('LOCAL_RANK', self.rank))
                device = torch.device(
                    'cuda', local % torch.cuda.device_count())
                torch.cuda.set_device(device)
            else:
                device = torch.device('cpu')
        self.device = device
        # Dedicated comm stream: collectives overlap backward compute
        # (HIP stream; None on CPU).
        self.comm_stream = (torch.cuda.Stream(device=device)
                            if device.type == 'cuda' else None)

    # -- basics -------------------------------------------------------------
    def barrier(self):
        if self.world_size > 1:
            dist.barrier()

    def all_reduce_(self, tensor, op=dist.ReduceOp.SUM, async_op=False):
        if self.world_size == 1:
            return None
        return dist.all_reduce(tensor, op=op, async_op=async_op)

    def broadcast_(self, tensor, src=0):
        if self.world_size > 1:
            dist.broadcast(tensor, src=src)

    def broadcast_parameters(self, named_params, src=0):
        """Rank-0 weight broadcast at init (contract N5; reference
        hvd.broadcast of all globals, mpi/graph_transform.py:26-32).
        Deterministic order: sorted by name (reference sorted broadcast,
        hybrid/runner.py:142-149)."""
        if self.world_size == 1:
            return
        by_name = dict(named_params)
        for name in sorted(by_name):
            dist.broadcast(by_name[name].detach(), src=src)

    # -- variable-length collectives -----------------------------------------
    def all_gather_v(self, tensor):
        """Allgather tensors whose dim-0 differs per rank (contract N4:
        Horovod's IndexedSlices allgather, mpi/graph_transform.py:46-61).
        Returns list of per-rank tensors."""
        if self.world_size == 1:
            return [tensor]
        n_local = torch.tensor([tensor.shape[0]], dtype=torch.long,
                               device=tensor.device)
        counts = [torch.zeros_like(n_local) for _ in range(self.world_size)]
        dist.all_gather(counts, n_local)
        counts = [int(c.item()) for c in counts]
        max_n = max(counts) if counts else 0
        tail_shape = tuple(tensor.shape[1:])
        padded = tensor.new_zeros((max_n,) + tail_shape)
        if tensor.shape[0] > 0:
            padded[:tensor.shape[0]] = tensor
        outs = [torch.empty_like(padded) for _ in range(self.world_size)]
        dist.all_gather(outs, padded)
        return [outs[r][:counts[r]] for r in range(self.world_size)]

    def all_gather_v_multi(self, tensors):
        """Allgather SEVERAL tensors that share dim-0 length per rank
        (ids/rows/counts of one sparse grad) with a SINGLE counts exchange.
        Returns a list (per input tensor) of lists (per rank)."""
        if self.world_size == 1:
            return [[t] for t in tensors]
        n_local = torch.tensor([tensors[0].shape[0]], dtype=torch.long,
                               device=tensors[0].device)
        counts_t = [torch.zeros_like(n_local)
                    for _ in range(self.world_size)]
        dist.all_gather(counts_t, n_local)
        counts = [int(c.item()) for c in counts_t]
        max_n = max(counts) if counts else 0
        outs_all = []
        for t in tensors:
            assert t.shape[0] == int(n_local.item())
            tail_shape = tuple(t.shape[1:])
            padded = t.new_zeros((max_n,) + tail_shape)
            if t.shape[0] > 0:
                padded[:t.shape[0]] = t
            outs = [torch.empty_like(padded)
                    for _ in range(self.world_size)]
            dist.all_gather(outs, padded)
            outs_all.append([outs[r][:counts[r]]
                             for r in range(self.world_size)])
        return outs_all

    def all_gather_fixed(self, tensor):
        """Equal-shape allgather, concatenated along dim 0 — the
        sync-free exchange of the padded sparse path (every rank
        contributes exactly the same count, so no counts exchange and no
        padding).  Returns a [world * n0, ...] tensor."""
        if self.world_size == 1:
            return tensor
        t = tensor.contiguous()
        out = t.new_empty((self.world_size * t.shape[0],)
                          + tuple(t.shape[1:]))
        if self.backend == 'nccl':
            dist.all_gather_into_tensor(out, t)
        else:
            chunks = list(out.chunk(self.world_size, dim=0))
            dist.all_gather(chunks, t)
            for r, c in enumerate(chunks):  # gloo may return new tensors
                start = r * t.shape[0]
                if not c.data_ptr() == out[start:start
                                           + t.shape[0]].data_ptr():
                    out[start:start + t.shape[0]] = c
        return out

    def all_to_all_v(self, tensors_per_rank):
        """Send tensors_per_rank[r] to rank r; receive one tensor from each
        rank (contract N7 PS transport).  Works on nccl (all_to_all) and
        gloo (allgather emulation for the CPU plumbing tests)."""
        ws = self.world_size
        if ws == 1:
            return [tensors_per_rank[0]]
        if self.backend == 'nccl':
            # exchange counts, then all_to_all_single with splits
            in_counts = torch.tensor(
                [t.shape[0] for t in tensors_per_rank], dtype=torch.long,
                device=self.device)
            out_counts = torch.empty_like(in_counts)
            dist.all_to_all_single(out_counts, in_counts)
            tail_shape = tuple(tensors_per_rank[0].shape[1:])
            # explicit column count: reshape(0, -1) on a zero-row chunk
            # (a batch touching none of some remote shard's rows) is
            # illegal, so never infer with -1
            cols = 1
            for d in tail_shape:
                cols *= int(d)
            send = torch.cat([t.reshape(t.shape[0], cols)
                              for t in tensors_per_rank], dim=0)
            out_counts_l = [int(c) for c in out_counts.tolist()]
            in_counts_l = [int(c) for c in in_counts.tolist()]
            recv = send.new_empty((sum(out_counts_l), cols))
            dist.all_to_all_single(recv, send,
                                   output_split_sizes=out_counts_l,
                                   input_split_sizes=in_counts_l)
            outs, off = [], 0
            for c in out_counts_l:
                outs.append(recv[off:off + c].reshape((c,) + tail_shape))
                off += c
            return outs
        # gloo fallback: pairwise isend/irecv (counts via one allgather,
        # then point-to-point payloads — O(world) messages, not O(world)
        # full allgathers)
        my_counts = torch.tensor([t.shape[0] for t in tensors_per_rank],
                                 dtype=torch.long)
        all_counts = [torch.zeros_like(my_counts) for _ in range(ws)]
        dist.all_gather(all_counts, my_counts)
        tail_shape = tuple(tensors_per_rank[0].shape[1:])
        dtype = tensors_per_rank[0].dtype
        outs = [None] * ws
        reqs = []
        for src in range(ws):
            if src == self.rank:
                outs[src] = tensors_per_rank[src]
                continue
            n = int(all_counts[src][self.rank])
            outs[src] = torch.empty((n,) + tail_shape, dtype=dtype)
            if n > 0:
                reqs.append(dist.irecv(outs[src], src=src))
        for dst in range(ws):
            if dst == self.rank:
                continue
            t = tensors_per_rank[dst].contiguous()
            if t.shape[0] > 0:
                reqs.append(dist.isend(t, dst=dst))
        for r in reqs:
            r.wait()
        return outs

    # -- a2a convenience for (ids, rows) pairs -------------------------------
    def exchange_rows(self, ids_per_rank, rows_per_rank):
        recv_ids = self.all_to_all_v([i.unsqueeze(1) for i in ids_per_rank])
        recv_rows = self.all_to_all_v(rows_per_rank)
        return ([i.squeeze(1) for i in recv_ids], recv_rows)
