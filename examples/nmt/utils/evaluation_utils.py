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
