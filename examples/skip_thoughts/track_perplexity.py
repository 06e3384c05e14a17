"""Held-out perplexity evaluation for a trained skip-thoughts model.

Reference: examples/skip_thoughts/track_perplexity.py — periodically
evaluates sum(losses)/sum(weights) over an eval set and reports
perplexity = exp(mean per-word loss).  Here the decoders are scored
with a FULL softmax over the vocabulary (the training path uses sampled
softmax; evaluation wants the exact likelihood).
"""

import math
import os
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))


def _decoder_nll(model, dec, thought, tgt):
    """Sum of per-token NLL + token count for one decoder."""
    emb = model.emb(tgt[:, :-1])
    B, T, _ = emb.shape
    th = thought.unsqueeze(1).expand(B, T, thought.shape[-1])
    outs, _ = dec(torch.cat([emb, th], dim=-1))
    h = model.out_proj(outs).reshape(B * T, -1)
    logits = h @ model.emb.full_weight().t()
    y = tgt[:, 1:].reshape(-1)
    return F.cross_entropy(logits, y, reduction='sum'), y.numel()


@torch.no_grad()
def evaluate_perplexity(model, batches):
    """batches: iterable of (cur, prev, next) id tensors.  Returns
    (perplexity, total_tokens)."""
    model.eval()
    tot_nll, tot_tok = 0.0, 0
    for cur, prev, nxt in batches:
        _, thought = model.encoder(model.emb(cur))
        for dec, tgt in ((model.dec_prev, prev), (model.dec_next, nxt)):
            nll, n = _decoder_nll(model, dec, thought, tgt)
            tot_nll += float(nll)
            tot_tok += n
    return math.exp(tot_nll / max(tot_tok, 1)), tot_tok


def main():
    from parallax_amd.models.skip_thoughts import (SkipThoughts,
                                                   synthetic_batch)
    torch.manual_seed(0)
    model = SkipThoughts(vocab=512, emb_dim=32, units=64, num_sampled=64,
                         min_partitions=2)
    batches = [synthetic_batch(8, 12, vocab=512) for _ in range(4)]
    ppl, n = evaluate_perplexity(model, batches)
    print('perplexity = %.2f over %d tokens' % (ppl, n))


if __name__ == '__main__':
    main()
