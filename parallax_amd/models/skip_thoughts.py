"""skip_thoughts — GRU sentence encoder/decoder with Adam.

Reference: examples/skip_thoughts (~1.5k LoC): GRU encoder over word
embeddings, two GRU decoders reconstructing the previous/next sentence,
Adam optimizer, in-graph sharding via
``shard.create_num_shards_and_shard_id()``
(skip_thoughts/ops/input_ops.py:92-101).
Here: same architecture with a PartitionedEmbedding vocabulary (sparse
plane) and dense GRU/projection params (all-reduce plane).
"""

import torch
import torch.nn as nn

from parallax_amd import optim
from parallax_amd.ops.embedding import PartitionedEmbedding
from parallax_amd.ops.sampled_softmax import sampled_softmax_loss
from parallax_amd.partitions import get_partitioner
from parallax_amd.session import TrainGraph

VOCAB = 20000


class GRUCell(nn.Module):
    def __init__(self, input_size, hidden):
        super().__init__()
        self.hidden = hidden
        s = (input_size + hidden) ** -0.5
        self.w_x = nn.Parameter(
            torch.empty(input_size, 3 * hidden).uniform_(-s, s))
        self.w_h = nn.Parameter(
            torch.empty(hidden, 3 * hidden).uniform_(-s, s))
        self.b = nn.Parameter(torch.zeros(3 * hidden))

    def forward(self, x, h):
        from parallax_amd.ops.gru import gru_gates
        xg = x @ self.w_x + self.b
        hg = h @ self.w_h
        return gru_gates(xg, hg, h)


class GRU(nn.Module):
    def __init__(self, input_size, hidden):
        super().__init__()
        self.cell = GRUCell(input_size, hidden)
        self.hidden = hidden

    def forward(self, xs, h=None):
        B, T, _ = xs.shape
        if h is None:
            h = xs.new_zeros(B, self.hidden)
        outs = []
        for t in range(T):
            h = self.cell(xs[:, t], h)
            outs.append(h)
        return torch.stack(outs, dim=1), h


class SkipThoughts(nn.Module):
    def __init__(self, vocab=VOCAB, emb_dim=256, units=512,
                 num_sampled=512, min_partitions=4):
        super().__init__()
        part = get_partitioner(min_partitions)
        self.emb = PartitionedEmbedding(vocab, emb_dim, part,
                                        init_scale=0.08)
        self.encoder = GRU(emb_dim, units)
        self.dec_prev = GRU(emb_dim + units, units)
        self.dec_next = GRU(emb_dim + units, units)
        self.out_proj = nn.Linear(units, emb_dim)
        self.vocab = vocab
        self.num_sampled = num_sampled

    def _decode_loss(self, dec, thought, tgt):
        emb = self.emb(tgt[:, :-1])
        B, T, E = emb.shape
        th = thought.unsqueeze(1).expand(B, T, thought.shape[-1])
        outs, _ = dec(torch.cat([emb, th], dim=-1))
        h = self.out_proj(outs).reshape(B * T, -1)
        y = tgt[:, 1:].reshape(-1)
        return sampled_softmax_loss(self.emb, None, y, h,
                                    self.num_sampled, self.vocab).mean()

    def forward(self, cur, prev, nxt):
        _, thought = self.encoder(self.emb(cur))
        return (self._decode_loss(self.dec_prev, thought, prev) +
                self._decode_loss(self.dec_next, thought, nxt))


def build_skip_thoughts_graph(vocab=VOCAB, emb_dim=256, units=512,
                              num_sampled=512, min_partitions=4, lr=8e-4,
                              use_bf16=True):
    model = SkipThoughts(vocab, emb_dim, units, num_sampled, min_partitions)
    optimizer = optim.Adam(model.parameters(), lr=lr)

    def step_fn(m, batch):
        cur, prev, nxt = batch
        if use_bf16 and cur.is_cuda:
            with torch.autocast('cuda', dtype=torch.bfloat16):
                loss = m(cur, prev, nxt)
        else:
            loss = m(cur, prev, nxt)
        return loss, {}

    return TrainGraph(model, optimizer, step_fn)


def synthetic_batch(batch_size=64, seq_len=20, vocab=VOCAB, device='cpu'):
    mk = lambda: torch.randint(0, vocab, (batch_size, seq_len),
                               device=device)
    return mk(), mk(), mk()
