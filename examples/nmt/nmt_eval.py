"""NMT eval driver: greedy decoding + corpus BLEU over a parallel dev
set (reference ``examples/nmt``'s external_eval / evaluation_utils
path, re-designed as a plain script over the MI355X-native model).

Usage:
  python nmt_eval.py --src dev.src --tgt dev.tgt \
      --src-vocab vocab.src --tgt-vocab vocab.tgt --ckpt ckpt_dir
With no files given it runs a self-contained smoke: builds vocab from a
tiny inline corpus, decodes with a random-init model and reports BLEU
(near 0 — the point is exercising the pipeline end to end).
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import torch

from parallax_amd.models.nmt import NMTTransformer
from utils import evaluation_utils, iterator_utils, vocab_utils


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--src')
    ap.add_argument('--tgt')
    ap.add_argument('--src-vocab')
    ap.add_argument('--tgt-vocab')
    ap.add_argument('--ckpt')
    ap.add_argument('--batch-size', type=int, default=32)
    ap.add_argument('--max-decode-len', type=int, default=64)
    ap.add_argument('--arch', default='transformer',
                    choices=['transformer', 'gnmt'])
    ap.add_argument('--layers', type=int, default=2)
    ap.add_argument('--d-model', type=int, default=64)
    ap.add_argument('--heads', type=int, default=4)
    ap.add_argument('--ffn', type=int, default=128)
    ap.add_argument('--beam-width', type=int, default=0,
                    help='0 = greedy (reference nmt.py beam_width flag)')
    ap.add_argument('--length-penalty', type=float, default=0.0)
    ap.add_argument('--metric', default='bleu',
                    choices=['bleu', 'rouge', 'accuracy', 'word_accuracy'])
    ap.add_argument('--subword', default=None, choices=['bpe'],
                    help='de-BPE both sides before scoring')
    args = ap.parse_args()

    if args.src:
        with open(args.src) as f:
            src_lines = [ln.strip() for ln in f]
        with open(args.tgt) as f:
            tgt_lines = [ln.strip() for ln in f]
        sv = vocab_utils.Vocab.load(args.src_vocab)
        tv = vocab_utils.Vocab.load(args.tgt_vocab)
    else:  # smoke mode
        src_lines = ['a b c d', 'b c', 'd d a'] * 4
        tgt_lines = ['x y z', 'y y', 'z x'] * 4
        sv = vocab_utils.Vocab('a b c d'.split())
        tv = vocab_utils.Vocab('x y z'.split())

    if args.arch == 'gnmt':
        from parallax_amd.models.nmt import GNMTAttention
        model = GNMTAttention(src_vocab=len(sv), tgt_vocab=len(tv),
                              emb_dim=args.d_model, units=args.d_model,
                              min_partitions=2)
    else:
        model = NMTTransformer(
            src_vocab=len(sv), tgt_vocab=len(tv), d_model=args.d_model,
            heads=args.heads, ffn=args.ffn, enc_layers=args.layers,
            dec_layers=args.layers, min_partitions=2)
    if args.ckpt:
        from parallax_amd.checkpoint import latest_checkpoint
        state = torch.load(latest_checkpoint(args.ckpt),
                           map_location='cpu')
        model.load_state_dict(state['model'])
    model.eval()

    examples = iterator_utils.make_examples(src_lines, tgt_lines, sv, tv)
    refs, hyps = [], []
    for batch in iterator_utils.bucketed_batches(
            examples, args.batch_size, shuffle=False):
        src, _tgt_in, tgt_out, _sl, _tl = batch
        if args.beam_width > 0:
            out_ids = model.beam_decode(
                src, beam_width=args.beam_width,
                length_penalty_weight=args.length_penalty,
                max_len=args.max_decode_len)
        else:
            out_ids = model.greedy_decode(src,
                                          max_len=args.max_decode_len)
        for row, ref_row in zip(out_ids, tgt_out.tolist()):
            hyps.append(tv.decode(row))
            refs.append(tv.decode(ref_row))
    score = evaluation_utils.evaluate(refs, hyps, args.metric,
                                      subword=args.subword)
    print('sentences=%d %s=%.2f BLEU=%.2f' % (
        len(hyps), args.metric, 100.0 * score,
        100.0 * evaluation_utils.corpus_bleu(refs, hyps, smooth=True)))


if __name__ == '__main__':
    main()
