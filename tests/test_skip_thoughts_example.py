"""skip_thoughts example inference-layer tests (reference surface:
skip_thoughts_encoder.py / encoder_manager.py / vocabulary_expansion.py /
track_perplexity.py)."""

import os
import sys

import numpy as np
import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, 'examples', 'skip_thoughts'))

from parallax_amd.config import CheckPointConfig  # noqa: E402
from parallax_amd.checkpoint import CheckpointManager  # noqa: E402
from parallax_amd.models.skip_thoughts import SkipThoughts  # noqa: E402

import encoder as st_encoder  # noqa: E402
import track_perplexity  # noqa: E402
import vocabulary_expansion  # noqa: E402

KW = dict(vocab=64, emb_dim=16, units=24, num_sampled=32, min_partitions=2)
VOCAB = {w: i for i, w in enumerate(
    ['<unk>', 'the', 'cat', 'sat', 'mat', 'dog', 'ran'])}


def _save_ckpt(tmp_path, model):
    cfg = CheckPointConfig(ckpt_dir=str(tmp_path), save_ckpt_steps=1)
    CheckpointManager(cfg, model, None, is_chief=True).save(0)


def test_encoder_checkpoint_roundtrip(tmp_path):
    torch.manual_seed(0)
    model = SkipThoughts(**KW)
    _save_ckpt(tmp_path, model)
    enc = st_encoder.SkipThoughtsEncoder.from_checkpoint(
        str(tmp_path), VOCAB, **KW)
    live = st_encoder.SkipThoughtsEncoder(model, VOCAB)
    sents = ['the cat sat', 'the dog ran', ['mat'], 'unseen words here']
    a = enc.encode(sents)
    b = live.encode(sents)
    assert len(a) == 4
    for va, vb in zip(a, b):
        assert va.shape == (KW['units'],)
        assert np.allclose(va, vb, atol=1e-6)   # restored == live
        assert abs(np.linalg.norm(va) - 1.0) < 1e-5   # use_norm
    raw = enc.encode(sents, use_norm=False)
    assert not np.allclose(raw[0], a[0])


def test_encoder_padding_invariance():
    """Batched (padded) encoding matches one-by-one encoding — the
    final state is taken at each sentence's true length."""
    torch.manual_seed(1)
    enc = st_encoder.SkipThoughtsEncoder(SkipThoughts(**KW), VOCAB)
    sents = ['the', 'the cat sat mat dog ran cat sat', 'dog ran']
    together = enc.encode(sents)
    alone = [enc.encode([s])[0] for s in sents]
    for va, vb in zip(together, alone):
        assert np.allclose(va, vb, atol=1e-5)


def test_encoder_expanded_embedding_feed():
    torch.manual_seed(2)
    model = SkipThoughts(**KW)
    emb_dict = {'zzz': np.ones(KW['emb_dim'], dtype=np.float32) * 0.1,
                'qqq': np.full(KW['emb_dim'], -0.2, dtype=np.float32)}
    enc = st_encoder.SkipThoughtsEncoder(model, VOCAB,
                                         embedding_dict=emb_dict)
    out = enc.encode(['zzz qqq', 'zzz'])
    assert len(out) == 2 and out[0].shape == (KW['units'],)
    assert not np.allclose(out[0], out[1])


def test_encoder_manager_concatenates(tmp_path):
    torch.manual_seed(3)
    m1, m2 = SkipThoughts(**KW), SkipThoughts(**KW)
    mgr = st_encoder.EncoderManager()
    with pytest.raises(ValueError):
        mgr.encode(['x'])
    mgr.add_encoder(st_encoder.SkipThoughtsEncoder(m1, VOCAB))
    mgr.add_encoder(st_encoder.SkipThoughtsEncoder(m2, VOCAB))
    out = mgr.encode(['the cat', 'dog'])
    assert len(out) == 2
    assert out[0].shape == (2 * KW['units'],)
    solo = st_encoder.SkipThoughtsEncoder(m1, VOCAB).encode(['the cat'])
    assert np.allclose(out[0][:KW['units']], solo[0], atol=1e-6)


def test_vocabulary_expansion_linear_map(tmp_path):
    rng = np.random.RandomState(0)
    D_w2v, D_st = 4, 3
    A = rng.randn(D_w2v, D_st)
    shared = ['w%d' % i for i in range(20)]
    w2v_vocab = {w: i for i, w in enumerate(
        shared + ['new1', 'two_words'])}
    w2v_emb = rng.randn(len(w2v_vocab), D_w2v)
    st_vocab = {w: i for i, w in enumerate(shared)}
    # trained vectors are EXACTLY linear in w2v (and 20 samples >> 4
    # features) -> the regression recovers A
    st_emb = w2v_emb[:len(shared)] @ A
    comb = vocabulary_expansion.expand_vocabulary(
        st_emb, st_vocab, w2v_vocab, w2v_emb)
    assert 'two_words' not in comb            # '_' words skipped
    for w in st_vocab:                        # trained vectors win
        assert np.allclose(comb[w], st_emb[st_vocab[w]])
    want = w2v_emb[w2v_vocab['new1']] @ A     # regressed new word
    assert np.allclose(comb['new1'], want, atol=1e-6)
    # load_vocabulary
    p = tmp_path / 'v.txt'
    p.write_text('a\nb\nc\n')
    assert vocabulary_expansion.load_vocabulary(str(p)) == \
        {'a': 0, 'b': 1, 'c': 2}


def test_perplexity_of_untrained_model_near_vocab_size():
    from parallax_amd.models.skip_thoughts import synthetic_batch
    torch.manual_seed(4)
    model = SkipThoughts(**KW)
    batches = [synthetic_batch(8, 10, vocab=KW['vocab'])
               for _ in range(3)]
    ppl, n = track_perplexity.evaluate_perplexity(model, batches)
    assert n == 3 * 8 * 9 * 2
    # untrained near-uniform predictions: ppl ~= vocab size
    assert 0.5 * KW['vocab'] < ppl < 2.0 * KW['vocab'], ppl
