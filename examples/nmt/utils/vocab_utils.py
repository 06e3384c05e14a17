"""Vocabulary + subword (BPE) utilities for the NMT example.

Parity targets (reference ``examples/nmt/utils/vocab_utils.py`` — vocab
files with <unk>/<s>/</s> reserved at ids 0/1/2 and lookup with an UNK
default — plus the BPE-merges application the reference assumes was done
offline by ``apply_bpe`` before training).  Re-designed as plain Python:
no TF lookup tables; a Vocab object with O(1) dict lookup feeds the
bucketed iterator directly.
"""

import collections
import os

UNK = '<unk>'
SOS = '<s>'
EOS = '</s>'
UNK_ID = 0
SOS_ID = 1
EOS_ID = 2
_SPECIALS = (UNK, SOS, EOS)


class Vocab:
    """word <-> id with the reference's reserved specials at 0/1/2."""

    def __init__(self, words):
        self.itos = list(_SPECIALS) + [w for w in words
                                       if w not in _SPECIALS]
        self.stoi = {w: i for i, w in enumerate(self.itos)}

    def __len__(self):
        return len(self.itos)

    def lookup(self, word):
        return self.stoi.get(word, UNK_ID)

    def encode(self, tokens):
        return [self.lookup(t) for t in tokens]

    def decode(self, ids, strip_specials=True):
        toks = []
        for i in ids:
            if strip_specials and i in (SOS_ID, EOS_ID):
                if i == EOS_ID:
                    break
                continue
            toks.append(self.itos[i] if 0 <= i < len(self.itos) else UNK)
        return toks

    def save(self, path):
        with open(path, 'w') as f:
            for w in self.itos:
                f.write(w + '\n')

    @classmethod
    def load(cls, path):
        with open(path) as f:
            words = [ln.rstrip('\n') for ln in f if ln.rstrip('\n')]
        # a vocab file may or may not already carry the specials
        return cls([w for w in words if w not in _SPECIALS])


def build_vocab(corpus_paths, max_size=None, min_count=1):
    """Frequency-sorted vocab over whitespace-tokenized corpora."""
    counts = collections.Counter()
    for p in corpus_paths:
        with open(p) as f:
            for line in f:
                counts.update(line.split())
    words = [w for w, c in counts.most_common() if c >= min_count]
    if max_size is not None:
        words = words[:max(0, max_size - len(_SPECIALS))]
    return Vocab(words)


# ---------------------------------------------------------------------------
# Byte-pair encoding: learn merges on a corpus and apply them, producing
# the "sub@@ word" convention the reference's BPE'd corpora use.
# ---------------------------------------------------------------------------

_EOW = '</w>'


def learn_bpe(corpus_paths, num_merges):
    """Learn BPE merge operations (list of symbol pairs, most frequent
    first) from whitespace-tokenized corpora."""
    word_counts = collections.Counter()
    for p in corpus_paths:
        with open(p) as f:
            for line in f:
                word_counts.update(line.split())
    # each word as a tuple of symbols, last symbol marks end-of-word
    vocab = {tuple(w[:-1]) + (w[-1] + _EOW,): c
             for w, c in word_counts.items() if w}
    merges = []
    for _ in range(num_merges):
        pairs = collections.Counter()
        for sym, c in vocab.items():
            for a, b in zip(sym, sym[1:]):
                pairs[(a, b)] += c
        if not pairs:
            break
        best = max(pairs, key=lambda k: (pairs[k], k))
        merges.append(best)
        merged = best[0] + best[1]
        new_vocab = {}
        for sym, c in vocab.items():
            out = []
            i = 0
            while i < len(sym):
                if (i + 1 < len(sym) and sym[i] == best[0]
                        and sym[i + 1] == best[1]):
                    out.append(merged)
                    i += 2
                else:
                    out.append(sym[i])
                    i += 1
            new_vocab[tuple(out)] = c
        vocab = new_vocab
    return merges


def apply_bpe(tokens, merges):
    """Split tokens into learned subwords; non-final pieces get the
    '@@' continuation marker (the reference corpora's convention)."""
    ranks = {pair: i for i, pair in enumerate(merges)}
    out = []
    for w in tokens:
        if not w:
            continue
        sym = list(w[:-1]) + [w[-1] + _EOW]
        while len(sym) > 1:
            best, bi = None, None
            for i, pair in enumerate(zip(sym, sym[1:])):
                r = ranks.get(pair)
                if r is not None and (best is None or r < best):
                    best, bi = r, i
            if best is None:
                break
            sym[bi:bi + 2] = [sym[bi] + sym[bi + 1]]
        pieces = []
        for i, s in enumerate(sym):
            last = (i == len(sym) - 1)
            s = s[:-len(_EOW)] if s.endswith(_EOW) else s
            pieces.append(s if last else s + '@@')
        out.extend(p for p in pieces if p)
    return out


def restore_bpe(tokens):
    """Invert apply_bpe: join '@@'-continued pieces back into words."""
    words, cur = [], ''
    for t in tokens:
        if t.endswith('@@'):
            cur += t[:-2]
        else:
            words.append(cur + t)
            cur = ''
    if cur:
        words.append(cur)
    return words
