"""skip_thoughts data preparation: tokenized sentences -> (prev, cur,
next) id-triple batches.

Reference: examples/skip_thoughts/data/preprocess_dataset.py (frequency
vocabulary with EOS=0 / UNK=1 specials, consecutive-sentence triples,
max-length filtering, optional EOS append) and special_words.py.  The
reference serializes TFRecord shards; here triples batch straight into
padded id tensors for the training driver (no serialization step on a
288 GB single-node setup), with the worker shard filter applied at the
triple level via parallax_amd.shard.
"""

import collections
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

from parallax_amd import shard as shard_lib  # noqa: E402

# special_words.py: first two ids, fixed
EOS = '<eos>'
EOS_ID = 0
UNK = '<unk>'
UNK_ID = 1


def build_vocabulary(input_files, num_words=20000):
    """Frequency-ordered word -> id dict with EOS=0 / UNK=1
    (preprocess_dataset.py:89-136)."""
    counts = collections.Counter()
    for path in input_files:
        with open(path) as f:
            for line in f:
                counts.update(line.split())
    vocab = {EOS: EOS_ID, UNK: UNK_ID}
    for w, _ in counts.most_common():
        if len(vocab) >= num_words:
            break
        if w not in vocab:
            vocab[w] = len(vocab)
    return vocab


def sentence_to_ids(sentence, vocab, add_eos=True):
    """Tokenized sentence (list of words) -> ids, UNK for OOV, optional
    trailing EOS (preprocess_dataset.py:156-161)."""
    ids = [vocab.get(w, UNK_ID) for w in sentence]
    if add_eos:
        ids.append(EOS_ID)
    return ids


def make_triples(input_files, vocab, max_sentence_length=30, add_eos=True):
    """Consecutive-sentence (prev, cur, next) triples; a triple is kept
    only if all three sentences fit max_sentence_length BEFORE the EOS
    insert (preprocess_dataset.py:175-210).  Sentence sequence does not
    cross file boundaries."""
    triples = []
    for path in input_files:
        with open(path) as f:
            sents = [ln.split() for ln in f if ln.strip()]
        for i in range(1, len(sents) - 1):
            window = (sents[i - 1], sents[i], sents[i + 1])
            if max_sentence_length and any(
                    len(s) > max_sentence_length for s in window):
                continue
            triples.append(tuple(
                sentence_to_ids(s, vocab, add_eos) for s in window))
    return triples


def _pad(rows, pad_id=EOS_ID):
    L = max(len(r) for r in rows)
    out = torch.full((len(rows), L), pad_id, dtype=torch.long)
    for i, r in enumerate(rows):
        out[i, :len(r)] = torch.tensor(r, dtype=torch.long)
    return out


def triple_batches(triples, batch_size, shuffle=True, seed=0,
                   num_shards=None, shard_id=None):
    """Yield (cur, prev, next) padded id tensors over THIS worker's
    shard of the triple list (model argument order:
    SkipThoughts.forward(cur, prev, nxt)).  Trailing partial batches
    are kept."""
    import random
    mine = list(shard_lib.shard(triples, num_shards, shard_id))
    order = list(mine)
    if shuffle:
        random.Random(seed).shuffle(order)
    for lo in range(0, len(order), batch_size):
        chunk = order[lo:lo + batch_size]
        prev = _pad([t[0] for t in chunk])
        cur = _pad([t[1] for t in chunk])
        nxt = _pad([t[2] for t in chunk])
        yield cur, prev, nxt
