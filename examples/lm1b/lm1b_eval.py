"""LM1B held-out perplexity evaluation.

Reference: the lm1b example reports training log-perplexity through its
sampled-softmax loss (language_model.py:95-100); evaluation scores the
exact likelihood — a FULL softmax over the (partitioned) vocabulary —
over one pass of a token corpus, LSTM state carried across batch
windows within each sentence stream (the reference's continuous-stream
convention, data_utils.py batching).

Usage:
  python lm1b_eval.py --corpus eval.txt --vocab vocab.txt \
      [--ckpt ckpt_dir] [--batches N]
With no files it runs a self-contained smoke on the bundled testdata
(random-init model; the point is exercising the eval path end to end).
"""

import argparse
import math
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import torch

import data_utils
from parallax_amd.models.lm1b import LM1BModel


@torch.no_grad()
def evaluate_perplexity(model, dataset, batch_size=16, num_steps=20,
                        max_batches=20, device='cpu'):
    """Full-softmax perplexity over max_batches windows of the
    continuous sentence streams.  Returns (perplexity, tokens)."""
    model.eval()
    w_full = model.softmax_w.full_weight().to(device)
    b_full = model.softmax_b.to(device)
    it = dataset.iterate_forever(batch_size, num_steps)
    state = None
    tot_nll, tot_tok = 0.0, 0
    for _ in range(max_batches):
        x, y, w = next(it)
        x = torch.from_numpy(x).to(device)
        y = torch.from_numpy(y).to(device)
        emb = model.emb(x)
        body = model.lstm.w.dtype
        if emb.dtype != body:
            emb = emb.to(body)
        hs, state = model.lstm(emb, state)
        logits = (hs.reshape(-1, hs.shape[-1]).float()
                  @ w_full.t().float() + b_full.float())
        nll = torch.nn.functional.cross_entropy(
            logits, y.reshape(-1), reduction='sum')
        tot_nll += float(nll)
        tot_tok += y.numel()
        # detach the carried state between windows
        state = tuple(s.detach() for s in state)
    return math.exp(tot_nll / max(tot_tok, 1)), tot_tok


def main():
    ap = argparse.ArgumentParser()
    here = os.path.dirname(os.path.abspath(__file__))
    ap.add_argument('--corpus',
                    default=os.path.join(here, 'testdata',
                                         'test_sentences.txt'))
    ap.add_argument('--vocab',
                    default=os.path.join(here, 'testdata',
                                         'test_vocab.txt'))
    ap.add_argument('--ckpt', default=None)
    ap.add_argument('--batch-size', type=int, default=8)
    ap.add_argument('--num-steps', type=int, default=20)
    ap.add_argument('--batches', type=int, default=10)
    ap.add_argument('--emb-dim', type=int, default=64)
    ap.add_argument('--state-size', type=int, default=128)
    ap.add_argument('--proj-size', type=int, default=64)
    args = ap.parse_args()

    vocab = data_utils.Vocabulary(args.vocab)
    ds = data_utils.LM1BDataset(args.corpus, vocab)
    torch.manual_seed(0)
    model = LM1BModel(vocab_size=vocab.size, emb_dim=args.emb_dim,
                      state_size=args.state_size,
                      proj_size=args.proj_size,
                      num_sampled=max(2, vocab.size // 4),
                      min_partitions=2)
    if args.ckpt:
        from parallax_amd.checkpoint import latest_checkpoint
        payload = torch.load(latest_checkpoint(args.ckpt),
                             map_location='cpu', weights_only=True)
        model.load_state_dict(payload['model'])
    ppl, n = evaluate_perplexity(model, ds, args.batch_size,
                                 args.num_steps, args.batches)
    print('perplexity=%.2f tokens=%d vocab=%d' % (ppl, n, vocab.size))


if __name__ == '__main__':
    main()
