"""Vocabulary expansion for a trained skip-thoughts model.

Reference: examples/skip_thoughts/vocabulary_expansion.py — learn a
linear map from a large word2vec-style embedding space into the trained
RNN embedding space on the shared vocabulary (sklearn LinearRegression),
then synthesize RNN embeddings for every word2vec word; words seen in
training keep their trained vectors.

The word2vec side here is any (vocab: word->index, emb: [V, D] array)
pair — there is no gensim dependency (and no network for pretrained
vectors); callers load whatever vector file they have into that shape.
"""

import collections

import numpy as np


def load_vocabulary(path):
    """word-per-line file -> OrderedDict word -> 0-based id
    (vocabulary_expansion.py:101-118)."""
    vocab = collections.OrderedDict()
    with open(path) as f:
        for i, line in enumerate(f):
            w = line.strip()
            if w:
                vocab[w] = i
    return vocab


def expand_vocabulary(skip_thoughts_emb, skip_thoughts_vocab,
                      word2vec_vocab, word2vec_emb):
    """vocabulary_expansion.py:121-167 semantics.

    Args:
      skip_thoughts_emb: [V_st, D_st] trained embedding matrix.
      skip_thoughts_vocab: dict word -> row in skip_thoughts_emb.
      word2vec_vocab: dict word -> row in word2vec_emb.
      word2vec_emb: [V_w2v, D_w2v] external embedding matrix.

    Returns: OrderedDict word -> np.ndarray[D_st]; every word2vec word
    without an underscore gets a regressed vector, every trained word
    keeps its trained vector (trained vectors win on overlap).
    """
    import sklearn.linear_model
    skip_thoughts_emb = np.asarray(skip_thoughts_emb)
    word2vec_emb = np.asarray(word2vec_emb)
    shared = [w for w in word2vec_vocab if w in skip_thoughts_vocab]
    if not shared:
        raise ValueError('no shared words between the two vocabularies')
    shared_st = skip_thoughts_emb[[skip_thoughts_vocab[w] for w in shared]]
    shared_w2v = word2vec_emb[[word2vec_vocab[w] for w in shared]]
    model = sklearn.linear_model.LinearRegression()
    model.fit(shared_w2v, shared_st)
    combined = collections.OrderedDict()
    for w, i in word2vec_vocab.items():
        if '_' not in w:
            combined[w] = model.predict(
                word2vec_emb[i].reshape(1, -1)).reshape(-1)
    for w, i in skip_thoughts_vocab.items():
        combined[w] = skip_thoughts_emb[i]
    return combined
