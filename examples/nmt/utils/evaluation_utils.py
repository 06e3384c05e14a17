"""BLEU evaluation for the NMT example.

Parity target: reference ``examples/nmt/utils/evaluation_utils.py``
(corpus BLEU with brevity penalty, the tutorial-NMT "bleu" metric).
Standard BLEU-4: geometric mean of clipped n-gram precisions times the
brevity penalty, with optional +1 smoothing for short corpora.
"""

import collections
import math


def _ngrams(tokens, n):
    return collections.Counter(
        tuple(tokens[i:i + n]) for i in range(len(tokens) - n + 1))


def corpus_bleu(references, hypotheses, max_order=4, smooth=False):
    """references / hypotheses: lists of token lists (one per sentence)."""
    matches = [0] * max_order
    possible = [0] * max_order
    ref_len = hyp_len = 0
    for ref, hyp in zip(references, hypotheses):
        ref_len += len(ref)
        hyp_len += len(hyp)
        for n in range(1, max_order + 1):
            ref_ng = _ngrams(ref, n)
            hyp_ng = _ngrams(hyp, n)
            overlap = sum((ref_ng & hyp_ng).values())
            matches[n - 1] += overlap
            possible[n - 1] += max(0, len(hyp) - n + 1)
    precisions = []
    for n in range(max_order):
        if smooth:
            p = (matches[n] + 1.0) / (possible[n] + 1.0)
        else:
            p = matches[n] / possible[n] if possible[n] > 0 else 0.0
        precisions.append(p)
    if min(precisions) <= 0:
        return 0.0
    log_p = sum(math.log(p) for p in precisions) / max_order
    bp = 1.0 if hyp_len > ref_len else math.exp(
        1.0 - ref_len / max(1, hyp_len))
    return bp * math.exp(log_p)


# --- additional reference metrics (evaluation_utils.py:31-64 dispatch:
# bleu / rouge / accuracy / word_accuracy) --------------------------------

def _lcs_len(x, y):
    """Longest-common-subsequence length (O(nm) DP)."""
    prev = [0] * (len(y) + 1)
    for xi in x:
        cur = [0]
        for j, yj in enumerate(y, 1):
            cur.append(prev[j - 1] + 1 if xi == yj
                       else max(prev[j], cur[-1]))
        prev = cur
    return prev[-1]


def rouge_l(references, hypotheses):
    """Mean sentence-level ROUGE-L F-score (scripts/rouge.py:112-230
    convention: beta = p/r, f = (1+b^2)pr / (r + b^2 p))."""
    total, n = 0.0, 0
    for ref, hyp in zip(references, hypotheses):
        llcs = _lcs_len(ref, hyp)
        r = llcs / len(ref) if ref else 0.0
        p = llcs / len(hyp) if hyp else 0.0
        beta = p / (r + 1e-12)
        denom = r + beta * beta * p
        total += ((1 + beta * beta) * r * p / denom) if denom > 0 else 0.0
        n += 1
    return total / max(n, 1)


def accuracy(references, hypotheses):
    """Fraction of exactly matching sentences (evaluation_utils
    _accuracy, as a 0-1 fraction)."""
    match = sum(1 for r, h in zip(references, hypotheses) if r == h)
    return match / max(len(references), 1)


def word_accuracy(references, hypotheses):
    """Mean per-sentence positional word accuracy (evaluation_utils
    _word_accuracy, as a 0-1 fraction)."""
    total = 0.0
    for ref, hyp in zip(references, hypotheses):
        m = sum(1 for a, b in zip(ref, hyp) if a == b)
        total += m / max(len(ref), len(hyp), 1)
    return total / max(len(references), 1)


def evaluate(references, hypotheses, metric='bleu', subword=None):
    """Metric dispatch over token-list corpora (reference
    evaluation_utils.evaluate works on files; callers here already hold
    token lists).  subword='bpe' de-BPEs both sides before scoring
    (reference subword_option)."""
    if subword == 'bpe':
        from .vocab_utils import restore_bpe
        references = [restore_bpe(r) for r in references]
        hypotheses = [restore_bpe(h) for h in hypotheses]
    elif subword:
        raise ValueError('unknown subword option %r' % (subword,))
    if metric == 'bleu':
        return corpus_bleu(references, hypotheses, smooth=True)
    if metric == 'rouge':
        return rouge_l(references, hypotheses)
    if metric == 'accuracy':
        return accuracy(references, hypotheses)
    if metric == 'word_accuracy':
        return word_accuracy(references, hypotheses)
    raise ValueError('unknown metric %r' % (metric,))
