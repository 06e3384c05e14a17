"""Downstream classification evaluation for skip-thoughts encodings.

Reference: examples/skip_thoughts/evaluate.py, which delegates the
MR/CR/SUBJ/MPQA tasks to the skip-thoughts paper's
``eval_classification.eval_nested_kfold`` — logistic regression over
the sentence encodings with nested k-fold cross-validation (inner folds
pick the L2 strength, outer folds report accuracy).  The task datasets
are not bundled (no network); ``eval_nested_kfold`` takes any
(sentences, binary/multiclass labels) pair, and the driver runs a
self-contained smoke on synthetic data.
"""

import argparse
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))


def eval_nested_kfold(encoder, sentences, labels, k=10,
                      cs=(0.01, 0.1, 1.0, 10.0), seed=0):
    """Nested k-fold logistic-regression accuracy over encodings.

    encoder: SkipThoughtsEncoder/EncoderManager (.encode(list) -> list
    of vectors).  Returns (mean outer-fold accuracy, per-fold list).
    """
    from sklearn.linear_model import LogisticRegression
    from sklearn.model_selection import StratifiedKFold, cross_val_score
    X = np.stack(encoder.encode(sentences))
    y = np.asarray(labels)
    k = min(k, np.bincount(y).min(), len(y))
    if k < 2:
        raise ValueError('need >= 2 examples per class')
    outer = StratifiedKFold(n_splits=k, shuffle=True, random_state=seed)
    accs = []
    for tr, te in outer.split(X, y):
        best_c, best = None, -1.0
        inner_k = min(3, np.bincount(y[tr]).min())
        for c in cs:
            clf = LogisticRegression(C=c, max_iter=1000)
            s = cross_val_score(clf, X[tr], y[tr], cv=inner_k).mean() \
                if inner_k >= 2 else 0.0
            if s > best:
                best, best_c = s, c
        clf = LogisticRegression(C=best_c, max_iter=1000)
        clf.fit(X[tr], y[tr])
        accs.append(float(clf.score(X[te], y[te])))
    return float(np.mean(accs)), accs


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--ckpt_dir', default=None,
                    help='trained checkpoint; default: random-init smoke')
    ap.add_argument('--sentences', default=None,
                    help='one tokenized sentence per line')
    ap.add_argument('--labels', default=None,
                    help='one integer label per line')
    ap.add_argument('--folds', type=int, default=10)
    args = ap.parse_args()

    import encoder as st_encoder
    from parallax_amd.models.skip_thoughts import SkipThoughts
    import torch
    torch.manual_seed(0)

    if args.sentences:
        with open(args.sentences) as f:
            sentences = [ln.strip() for ln in f if ln.strip()]
        with open(args.labels) as f:
            labels = [int(ln) for ln in f if ln.strip()]
        words = sorted({w for s in sentences for w in s.split()})
    else:   # synthetic smoke: two trivially separable "topics"
        rng = np.random.RandomState(0)
        sentences, labels = [], []
        for i in range(60):
            topic = i % 2
            base = ['cat', 'dog', 'pet'] if topic else ['stock', 'bank',
                                                        'trade']
            sentences.append(' '.join(rng.choice(base, 4)))
            labels.append(topic)
        words = ['cat', 'dog', 'pet', 'stock', 'bank', 'trade']
    vocab = {w: i for i, w in enumerate(words)}

    kw = dict(vocab=max(len(vocab), 8), emb_dim=32, units=64,
              num_sampled=8, min_partitions=2)
    if args.ckpt_dir:
        enc = st_encoder.SkipThoughtsEncoder.from_checkpoint(
            args.ckpt_dir, vocab, **kw)
    else:
        enc = st_encoder.SkipThoughtsEncoder(SkipThoughts(**kw), vocab)
    acc, folds = eval_nested_kfold(enc, sentences, labels, k=args.folds)
    print('nested %d-fold accuracy = %.4f (folds: %s)' %
          (len(folds), acc, ' '.join('%.2f' % a for a in folds)))


if __name__ == '__main__':
    main()
