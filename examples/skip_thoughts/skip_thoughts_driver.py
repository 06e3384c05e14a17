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
    ap.add_argument('--resource_info', default=os.path.join(
        os.path.dirname(os.path.abspath(__file__)), 'resource_info'))
    args = ap.parse_args()

    sess, num_workers, worker_id, _ = parallax_amd.parallel_run(
        lambda: st.build_skip_thoughts_graph(lr=args.learning_rate),
        args.resource_info,
        parallax_config=parallax_amd.Config(run_option='HYBRID'))
    device = sess.comm.device
    prev = time.time()
    for step in range(args.max_steps):
        out = sess.run(st.synthetic_batch(args.batch_size, args.seq_len,
                                          device=device))
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
