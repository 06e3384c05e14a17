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
    ap.add_argument('--resource_info', default=os.path.join(
        os.path.dirname(os.path.abspath(__file__)), 'resource_info'))
    args = ap.parse_args()

    sess, num_workers, worker_id, _ = parallax_amd.parallel_run(
        lambda: nmt.build_nmt_graph(arch=args.arch,
                                    lr=args.learning_rate),
        args.resource_info,
        parallax_config=parallax_amd.Config(run_option='HYBRID'))
    device = sess.comm.device
    prev = time.time()
    for step in range(args.max_steps):
        out = sess.run(nmt.synthetic_batch(args.batch_size, args.src_len,
                                           args.tgt_len, device=device))
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
