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


def test_lm1b_eval_driver_smoke():
    import subprocess
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, 'examples', 'lm1b',
                                      'lm1b_eval.py'), '--batches', '4'],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-1500:]
    assert 'perplexity=' in out.stdout


def test_lm1b_eval_perplexity_drops_with_training():
    """Full-softmax eval perplexity falls after a few training steps on
    the same corpus (untrained ~= vocab size)."""
    import math
    import importlib.util
    import torch
    spec = importlib.util.spec_from_file_location(
        'lm1b_eval', os.path.join(REPO, 'examples', 'lm1b',
                                  'lm1b_eval.py'))
    ev = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(ev)
    import data_utils
    from parallax_amd.models.lm1b import LM1BModel

    td = os.path.join(REPO, 'examples', 'lm1b', 'testdata')
    vocab = data_utils.Vocabulary(os.path.join(td, 'test_vocab.txt'))
    ds = data_utils.LM1BDataset(os.path.join(td, 'test_sentences.txt'),
                                vocab)
    torch.manual_seed(1)
    model = LM1BModel(vocab_size=vocab.size, emb_dim=16, state_size=32,
                      proj_size=16, num_sampled=8, min_partitions=2)
    before, _ = ev.evaluate_perplexity(model, ds, batch_size=4,
                                       num_steps=10, max_batches=5)
    assert 0.4 * vocab.size < before < 3.0 * vocab.size
    opt = torch.optim.Adagrad(model.parameters(), lr=0.1)
    it = ds.iterate_forever(4, 10)
    model.train()
    for _ in range(40):
        x, y, _w = next(it)
        loss, _st = model(torch.from_numpy(x), torch.from_numpy(y))
        opt.zero_grad()
        loss.backward()
        for p in model.parameters():
            if p.grad is not None and p.grad.is_sparse:
                p.grad = p.grad.to_dense()
        opt.step()
    after, _ = ev.evaluate_perplexity(model, ds, batch_size=4,
                                      num_steps=10, max_batches=5)
    assert after < before * 0.8, (before, after)
    assert not math.isnan(after)


def test_lm1b_eval_ckpt_restore(tmp_path):
    """lm1b_eval --ckpt restores a CheckpointManager-written checkpoint
    (dims via flags)."""
    import subprocess
    import torch
    from parallax_amd.config import CheckPointConfig
    from parallax_amd.checkpoint import CheckpointManager
    from parallax_amd.models.lm1b import LM1BModel
    td = os.path.join(REPO, 'examples', 'lm1b', 'testdata')
    vocab_lines = open(os.path.join(td, 'test_vocab.txt')).read()
    vsize = len([l for l in vocab_lines.splitlines() if l.strip()])
    torch.manual_seed(0)
    model = LM1BModel(vocab_size=vsize, emb_dim=16, state_size=32,
                      proj_size=16, num_sampled=4, min_partitions=2)
    cfg = CheckPointConfig(ckpt_dir=str(tmp_path), save_ckpt_steps=1)
    CheckpointManager(cfg, model, None, is_chief=True).save(5)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, 'examples', 'lm1b',
                                      'lm1b_eval.py'),
         '--ckpt', str(tmp_path), '--emb-dim', '16',
         '--state-size', '32', '--proj-size', '16', '--batches', '2'],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-1500:]
    assert 'perplexity=' in out.stdout


def test_lm1b_driver_real_data_smoke(tmp_path):
    import subprocess
    td = os.path.join(REPO, 'examples', 'lm1b', 'testdata')
    res_info = tmp_path / 'resource_info'
    res_info.write_text('127.0.0.1:\n')
    out = subprocess.run(
        [sys.executable,
         os.path.join(REPO, 'examples', 'lm1b',
                      'lm1b_distributed_driver.py'),
         '--batch_size', '4', '--num_steps', '6', '--max_steps', '2',
         '--data_path', os.path.join(td, 'test_sentences.txt'),
         '--vocab_path', os.path.join(td, 'test_vocab.txt'),
         '--resource_info', str(res_info)],
        capture_output=True, text=True, timeout=300,
        env=dict(os.environ, PYTHONPATH=REPO))
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
