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


def test_data_utils_vocab_and_triples(tmp_path):
    import preprocess_dataset as data_utils
    p = tmp_path / 'corpus.txt'
    p.write_text('a b c\n' 'b c d\n' 'c d\n' 'a very long sentence '
                 'with many many words beyond limit here ok\n' 'd a\n')
    vocab = data_utils.build_vocabulary([str(p)], num_words=8)
    assert vocab['<eos>'] == 0 and vocab['<unk>'] == 1
    assert len(vocab) == 8
    # triples over consecutive sentences, long ones filtered
    tr = data_utils.make_triples([str(p)], vocab, max_sentence_length=3)
    # windows: (0,1,2) ok; (1,2,3) has the long sentence; (2,3,4) too
    assert len(tr) == 1
    prev, cur, nxt = tr[0]
    assert prev[-1] == 0 and cur[-1] == 0 and nxt[-1] == 0   # EOS
    assert cur[:-1] == [vocab['b'], vocab['c'], vocab['d']]
    # OOV -> UNK
    assert data_utils.sentence_to_ids(['zzz'], vocab) == [1, 0]


def test_data_utils_batches_shard_and_pad(tmp_path):
    import preprocess_dataset as data_utils
    lines = '\n'.join('w%d w%d' % (i, i + 1) for i in range(30))
    p = tmp_path / 'c.txt'
    p.write_text(lines + '\n')
    vocab = data_utils.build_vocabulary([str(p)], num_words=40)
    tr = data_utils.make_triples([str(p)], vocab)
    assert len(tr) == 28
    got = 0
    for s in range(2):
        for cur, prev, nxt in data_utils.triple_batches(
                tr, 8, shuffle=False, num_shards=2, shard_id=s):
            assert cur.dtype == torch.long
            assert cur.shape == prev.shape == nxt.shape
            got += cur.shape[0]
    assert got == 28        # shards disjoint + complete, tails kept


def test_driver_real_data_smoke(tmp_path):
    import subprocess
    corpus = tmp_path / 'c.txt'
    corpus.write_text('\n'.join('tok%d tok%d tok%d' % (i % 7, (i + 1) % 7,
                                                       (i + 2) % 7)
                                for i in range(20)) + '\n')
    res_info = tmp_path / 'resource_info'
    res_info.write_text('127.0.0.1:\n')
    env = dict(os.environ, PYTHONPATH=REPO)
    out = subprocess.run(
        [sys.executable,
         os.path.join(REPO, 'examples', 'skip_thoughts',
                      'skip_thoughts_driver.py'),
         '--batch_size', '4', '--max_steps', '2',
         '--input_files', str(corpus), '--num_words', '16',
         '--resource_info', str(res_info)],
        capture_output=True, text=True, timeout=300, env=env)
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])


def test_driver_ckpt_feeds_encoder(tmp_path):
    """Driver trains with checkpointing; the saved checkpoint loads into
    the inference encoder (train -> encode round trip through the real
    stack)."""
    import subprocess
    corpus = tmp_path / 'c.txt'
    corpus.write_text('\n'.join('tok%d tok%d' % (i % 5, (i + 1) % 5)
                                for i in range(12)) + '\n')
    res_info = tmp_path / 'resource_info'
    res_info.write_text('127.0.0.1:\n')
    ckpt = str(tmp_path / 'ck')
    env = dict(os.environ, PYTHONPATH=REPO)
    out = subprocess.run(
        [sys.executable,
         os.path.join(REPO, 'examples', 'skip_thoughts',
                      'skip_thoughts_driver.py'),
         '--batch_size', '4', '--max_steps', '3',
         '--input_files', str(corpus), '--num_words', '8',
         '--ckpt_dir', ckpt, '--save_ckpt_steps', '2',
         '--resource_info', str(res_info)],
        capture_output=True, text=True, timeout=300, env=env)
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
    import preprocess_dataset as pd
    vocab = pd.build_vocabulary([str(corpus)], 8)
    enc = st_encoder.SkipThoughtsEncoder.from_checkpoint(
        ckpt, vocab, vocab=len(vocab), emb_dim=256, units=512,
        num_sampled=min(512, max(2, len(vocab) // 2)), min_partitions=4)
    vecs = enc.encode(['tok1 tok2', 'tok3'])
    assert len(vecs) == 2 and vecs[0].shape == (512,)


def test_eval_nested_kfold_separable_vs_shuffled():
    import numpy as np
    import evaluate as st_eval
    torch.manual_seed(5)
    enc = st_encoder.SkipThoughtsEncoder(SkipThoughts(**KW), VOCAB)
    rng = np.random.RandomState(0)
    sents, labels = [], []
    for i in range(40):
        topic = i % 2
        base = ['cat', 'dog', 'mat'] if topic else ['ran', 'sat', 'the']
        sents.append(' '.join(rng.choice(base, 4)))
        labels.append(topic)
    acc, folds = st_eval.eval_nested_kfold(enc, sents, labels, k=4)
    assert acc > 0.9, acc
    assert len(folds) == 4
    # shuffled labels: near chance
    sh = list(labels)
    rng.shuffle(sh)
    acc_sh, _ = st_eval.eval_nested_kfold(enc, sents, sh, k=4)
    assert acc_sh < 0.8, acc_sh
    with pytest.raises(ValueError):
        st_eval.eval_nested_kfold(enc, sents[:2], [0, 1], k=4)


def test_evaluate_driver_smoke():
    import subprocess
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, 'examples', 'skip_thoughts',
                                      'evaluate.py'), '--folds', '3'],
        capture_output=True, text=True, timeout=300,
        env=dict(os.environ, PYTHONPATH=REPO))
    assert out.returncode == 0, out.stderr[-1500:]
    assert 'accuracy' in out.stdout
