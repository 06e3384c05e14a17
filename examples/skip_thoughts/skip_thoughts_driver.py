"""skip_thoughts driver (reference examples/skip_thoughts): GRU
sentence encoder trained with Adam, in-graph sharding."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import parallax_amd
from parallax_amd.models import skip_thoughts as st


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--batch_size', type=int, default=64)
    ap.add_argument('--seq_len', type=int, default=20)
    ap.add_argument('--max_steps', type=int, default=1000000)
    ap.add_argument('--learning_rate', type=float, default=8e-4)
    ap.add_argument('--log_frequency', type=int, default=10)
    ap.add_argument('--input_files', default=None,
                    help='comma-separated tokenized sentence files '
                         '(reference preprocess_dataset input); '
                         'default: synthetic ids')
    ap.add_argument('--num_words', type=int, default=20000)
    ap.add_argument('--max_sentence_length', type=int, default=30)
    ap.add_argument('--run_option', default='HYBRID')
    ap.add_argument('--ckpt_dir', default=None)
    ap.add_argument('--save_ckpt_steps', type=int, default=None)
    ap.add_argument('--resource_info', default=os.path.join(
        os.path.dirname(os.path.abspath(__file__)), 'resource_info'))
    args = ap.parse_args()

    vocab_size = st.VOCAB
    triples = None
    if args.input_files:
        import preprocess_dataset as data_utils
        files = args.input_files.split(',')
        vocab = data_utils.build_vocabulary(files, args.num_words)
        triples = data_utils.make_triples(files, vocab,
                                          args.max_sentence_length)
        vocab_size = len(vocab)

    sess, num_workers, worker_id, _ = parallax_amd.parallel_run(
        lambda: st.build_skip_thoughts_graph(
            vocab=vocab_size, lr=args.learning_rate,
            num_sampled=min(512, max(2, vocab_size // 2))),
        args.resource_info,
        parallax_config=parallax_amd.Config(
            run_option=args.run_option,
            ckpt_config=parallax_amd.CheckPointConfig(
                ckpt_dir=args.ckpt_dir,
                save_ckpt_steps=args.save_ckpt_steps)))
    device = sess.comm.device

    def real_batches():
        import preprocess_dataset as data_utils
        epoch = 0
        while True:
            for b in data_utils.triple_batches(triples, args.batch_size,
                                               seed=epoch):
                yield tuple(t.to(device) for t in b)
            epoch += 1

    data_it = iter(real_batches()) if triples else None
    prev = time.time()
    for step in range(args.max_steps):
        batch = (next(data_it) if data_it is not None else
                 st.synthetic_batch(args.batch_size, args.seq_len,
                                    device=device))
        out = sess.run(batch)
        if step % args.log_frequency == 0 and worker_id == 0:
            now = time.time()
            parallax_amd.log.info('step %d, %.2f steps/sec, loss = %.4f',
                                  out['global_step'],
                                  args.log_frequency / max(now - prev,
                                                           1e-9),
                                  float(out['loss']))
            prev = now
    sess.close()


if __name__ == '__main__':
    main()
