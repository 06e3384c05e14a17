"""NMT distributed driver (reference examples/nmt/nmt_distributed_driver.py
:185-188): Transformer or GNMT-attention seq2seq, hybrid mode, partitioned
src/tgt embeddings, sharded synthetic input."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import parallax_amd
from parallax_amd.models import nmt


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--arch', default='transformer',
                    choices=['transformer', 'gnmt'])
    ap.add_argument('--batch_size', type=int, default=64)
    ap.add_argument('--src_len', type=int, default=48)
    ap.add_argument('--tgt_len', type=int, default=48)
    ap.add_argument('--max_steps', type=int, default=1000000)
    ap.add_argument('--learning_rate', type=float, default=1e-3)
    ap.add_argument('--log_frequency', type=int, default=10)
    ap.add_argument('--src', default=None,
                    help='tokenized source corpus (with --tgt: real-data '
                         'training through the bucketed iterator; '
                         'default: synthetic)')
    ap.add_argument('--tgt', default=None)
    ap.add_argument('--src_vocab', default=None,
                    help='vocab file; default: built from the corpus')
    ap.add_argument('--tgt_vocab', default=None)
    ap.add_argument('--vocab_size', type=int, default=32000)
    ap.add_argument('--src_max_len', type=int, default=50)
    ap.add_argument('--d_model', type=int, default=512)
    ap.add_argument('--heads', type=int, default=8)
    ap.add_argument('--ffn', type=int, default=2048)
    ap.add_argument('--layers', type=int, default=6)
    ap.add_argument('--run_option', default='HYBRID')
    ap.add_argument('--ckpt_dir', default=None)
    ap.add_argument('--save_ckpt_steps', type=int, default=None)
    ap.add_argument('--resource_info', default=os.path.join(
        os.path.dirname(os.path.abspath(__file__)), 'resource_info'))
    args = ap.parse_args()

    examples = sv = tv = None
    graph_kw = dict(d_model=args.d_model, heads=args.heads, ffn=args.ffn,
                    enc_layers=args.layers, dec_layers=args.layers)
    if args.src:
        from utils import iterator_utils, vocab_utils
        with open(args.src) as f:
            src_lines = [ln.strip() for ln in f if ln.strip()]
        with open(args.tgt) as f:
            tgt_lines = [ln.strip() for ln in f if ln.strip()]
        sv = (vocab_utils.Vocab.load(args.src_vocab) if args.src_vocab
              else vocab_utils.build_vocab([args.src], args.vocab_size))
        tv = (vocab_utils.Vocab.load(args.tgt_vocab) if args.tgt_vocab
              else vocab_utils.build_vocab([args.tgt], args.vocab_size))
        examples = iterator_utils.make_examples(
            src_lines, tgt_lines, sv, tv, src_max_len=args.src_max_len)
        graph_kw.update(src_vocab=len(sv), tgt_vocab=len(tv),
                        min_partitions=2)

    sess, num_workers, worker_id, _ = parallax_amd.parallel_run(
        lambda: nmt.build_nmt_graph(arch=args.arch,
                                    lr=args.learning_rate, **graph_kw),
        args.resource_info,
        parallax_config=parallax_amd.Config(
            run_option=args.run_option,
            ckpt_config=parallax_amd.CheckPointConfig(
                ckpt_dir=args.ckpt_dir,
                save_ckpt_steps=args.save_ckpt_steps)))
    device = sess.comm.device

    def real_batches():
        from utils import iterator_utils
        epoch = 0
        while True:
            for b in iterator_utils.bucketed_batches(
                    examples, args.batch_size, shuffle=True, seed=epoch,
                    shard_id=worker_id, num_shards=num_workers):
                src, tgt_in, tgt_out, _sl, _tl = b
                yield (src.to(device), tgt_in.to(device),
                       tgt_out.to(device))
            epoch += 1

    data_it = iter(real_batches()) if examples else None
    prev = time.time()
    for step in range(args.max_steps):
        batch = (next(data_it) if data_it is not None else
                 nmt.synthetic_batch(args.batch_size, args.src_len,
                                     args.tgt_len, device=device))
        out = sess.run(batch)
        if step % args.log_frequency == 0 and worker_id == 0:
            now = time.time()
            wps = (args.log_frequency * args.batch_size * args.tgt_len
                   * num_workers / max(now - prev, 1e-9))
            parallax_amd.log.info('step %d, tgt words/sec = %.0f, '
                                  'loss = %.4f', out['global_step'], wps,
                                  float(out['loss']))
            prev = now
    sess.close()


if __name__ == '__main__':
    main()
