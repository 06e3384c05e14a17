"""Length-bucketed batching for the NMT example.

Parity target: reference ``examples/nmt/utils/iterator_utils.py:109-173``
— filter by max_len, append EOS / prepend SOS, bucket by source length
(``bucket_width``), pad within a batch, and shard the stream across
workers with the framework's ``i % num_shards == shard_id`` semantics
(``parallax_amd.shard``).  Re-designed: a plain Python generator yielding
padded torch tensors; no TF Dataset graph.
"""

import random

import torch

from .vocab_utils import EOS_ID, SOS_ID


def make_examples(src_lines, tgt_lines, src_vocab, tgt_vocab,
                  src_max_len=None, tgt_max_len=None):
    """Tokenize + encode parallel text into (src_ids, tgt_ids) pairs,
    dropping empty and over-length sentences (reference behavior)."""
    out = []
    for s, t in zip(src_lines, tgt_lines):
        si = src_vocab.encode(s.split())
        ti = tgt_vocab.encode(t.split())
        if not si or not ti:
            continue
        if src_max_len is not None and len(si) > src_max_len:
            continue
        if tgt_max_len is not None and len(ti) > tgt_max_len:
            continue
        out.append((si, ti))
    return out


def _bucket_key(src_len, bucket_width):
    return src_len // max(1, bucket_width)


def bucketed_batches(examples, batch_size, bucket_width=5, shuffle=True,
                     seed=0, shard_id=0, num_shards=1, pad_id=EOS_ID,
                     drop_last=False):
    """Yield (src, tgt_in, tgt_out, src_len, tgt_len) padded LongTensors.

    * examples are sharded ``i % num_shards == shard_id`` (the framework's
      data-sharding contract, shard.py);
    * within a shard, examples group into buckets of similar src length
      so padding waste stays low;
    * tgt_in = SOS + tgt, tgt_out = tgt + EOS (teacher forcing).
    """
    rng = random.Random(seed)
    mine = [ex for i, ex in enumerate(examples)
            if i % num_shards == shard_id]
    if shuffle:
        rng.shuffle(mine)
    buckets = {}
    order = []
    for ex in mine:
        k = _bucket_key(len(ex[0]), bucket_width)
        b = buckets.setdefault(k, [])
        b.append(ex)
        if len(b) == batch_size:
            order.append(buckets.pop(k))
    if not drop_last:
        order.extend(b for b in buckets.values() if b)
    if shuffle:
        rng.shuffle(order)
    for batch in order:
        yield _pad_batch(batch, pad_id)


def _pad_batch(batch, pad_id):
    bs = len(batch)
    sl = max(len(s) for s, _ in batch)
    tl = max(len(t) for _, t in batch) + 1   # +1 for SOS/EOS
    src = torch.full((bs, sl), pad_id, dtype=torch.long)
    tgt_in = torch.full((bs, tl), pad_id, dtype=torch.long)
    tgt_out = torch.full((bs, tl), pad_id, dtype=torch.long)
    src_len = torch.zeros(bs, dtype=torch.long)
    tgt_len = torch.zeros(bs, dtype=torch.long)
    for i, (s, t) in enumerate(batch):
        src[i, :len(s)] = torch.tensor(s)
        tgt_in[i, 0] = SOS_ID
        tgt_in[i, 1:1 + len(t)] = torch.tensor(t)
        tgt_out[i, :len(t)] = torch.tensor(t)
        tgt_out[i, len(t)] = EOS_ID
        src_len[i] = len(s)
        tgt_len[i] = len(t) + 1
    return src, tgt_in, tgt_out, src_len, tgt_len
