"""LM1B data-utils tests (reference data_utils round-robin semantics +
the bundled tiny testdata, mirroring examples/lm1b/testdata)."""

import os
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, 'examples', 'lm1b'))

from data_utils import Vocabulary, LM1BDataset

TD = os.path.join(REPO, 'examples', 'lm1b', 'testdata')


def test_vocabulary():
    v = Vocabulary(os.path.join(TD, 'test_vocab.txt'))
    assert v.size == 16
    assert v.token_to_id('the') == 3
    assert v.token_to_id('NOTINVOCAB') == v.unk
    enc = v.encode('the lazy dog')
    assert enc[0] == v.bos and enc[-1] == v.eos
    assert v.id_to_token(enc[1]) == 'the'


def test_dataset_batches():
    v = Vocabulary(os.path.join(TD, 'test_vocab.txt'))
    ds = LM1BDataset(os.path.join(TD, 'test_sentences.txt'), v)
    it = ds.iterate_forever(batch_size=2, num_steps=5)
    x, y, w = next(it)
    assert x.shape == (2, 5) and y.shape == (2, 5)
    # y is x shifted by one within the stream
    x2, y2, _ = next(it)
    assert np.all(w == 1.0)
    assert x[0, 1] == y[0, 0]


def test_dataset_worker_round_robin():
    v = Vocabulary(os.path.join(TD, 'test_vocab.txt'))
    ds = LM1BDataset(os.path.join(TD, 'test_sentences.txt'), v)
    s0 = ds._sentence_stream(2, 0)
    s1 = ds._sentence_stream(2, 1)
    a = [next(s0) for _ in range(2)]
    b = [next(s1) for _ in range(2)]
    # worker 0 gets sentences 0,2; worker 1 gets 1,3
    assert a[0] == v.encode('the quick brown fox jumps over the lazy dog')
    assert b[0] == v.encode('a short sentence')
    assert a[1] == v.encode('another one')
    assert b[1] == v.encode('the lazy dog jumps')
