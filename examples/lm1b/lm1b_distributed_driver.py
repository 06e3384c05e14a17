"""LM1B distributed driver — the flagship hybrid workload.

MI355X-native counterpart of the reference's
examples/lm1b/lm1b_distributed_driver.py: LSTM LM over a 793,470-word
vocabulary, partitioned embeddings + sampled softmax on the sparse (PS)
plane, LSTM on the dense (all-reduce) plane, wps logging
(reference :98-106), synthetic-data mode (reference :81-84).
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import parallax_amd
from parallax_amd.models import lm1b


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--batch_size', type=int, default=128)
    ap.add_argument('--num_steps', type=int, default=20)
    ap.add_argument('--vocab_size', type=int, default=lm1b.VOCAB_SIZE)
    ap.add_argument('--max_steps', type=int, default=1000000)
    ap.add_argument('--learning_rate', type=float, default=0.2)
    ap.add_argument('--log_frequency', type=int, default=20)
    ap.add_argument('--use_synthetic', action='store_true', default=True)
    ap.add_argument('--data_path', default=None,
                    help='sentence corpus (with --vocab_path: real-data '
                         'training via the round-robin sentence streams; '
                         'disables --use_synthetic)')
    ap.add_argument('--vocab_path', default=None)
    ap.add_argument('--run_option', default='HYBRID')
    ap.add_argument('--ckpt_dir', default=None)
    ap.add_argument('--save_ckpt_steps', type=int, default=None)
    ap.add_argument('--profile_dir', default=None)
    ap.add_argument('--profile_worker', type=int, default=0)
    ap.add_argument('--cuda_graph', action='store_true', default=True)
    ap.add_argument('--pure_bf16', action='store_true', default=False,
                    help='bf16 LSTM body + fused fp32 master weights')
    ap.add_argument('--resource_info', default=os.path.join(
        os.path.dirname(os.path.abspath(__file__)), 'resource_info'))
    args = ap.parse_args()

    dataset = None
    if args.data_path:
        import data_utils
        vocab = data_utils.Vocabulary(args.vocab_path)
        dataset = data_utils.LM1BDataset(args.data_path, vocab)
        args.vocab_size = vocab.size
        args.use_synthetic = False

    config = parallax_amd.Config(
        run_option=args.run_option,
        ckpt_config=parallax_amd.CheckPointConfig(
            ckpt_dir=args.ckpt_dir, save_ckpt_steps=args.save_ckpt_steps),
        profile_config=parallax_amd.ProfileConfig(
            profile_dir=args.profile_dir, profile_range=(5, 8),
            profile_worker=args.profile_worker))

    sess, num_workers, worker_id, _ = parallax_amd.parallel_run(
        lambda: lm1b.build_lm1b_graph(batch_size=args.batch_size,
                                      num_steps=args.num_steps,
                                      vocab_size=args.vocab_size,
                                      lr=args.learning_rate,
                                      pure_bf16=args.pure_bf16),
        args.resource_info, parallax_config=config)
    device = sess.comm.device
    if args.cuda_graph and device.type == 'cuda' \
            and args.profile_dir is None:
        state0 = lm1b.make_init_state(args.batch_size, device=device,
                                      dtype=torch.bfloat16)
        sess.enable_cuda_graph(lm1b.synthetic_batch(
            args.batch_size, args.num_steps, args.vocab_size, device,
            state0))

    data_it = None
    if dataset is not None:
        data_it = dataset.iterate_forever(args.batch_size, args.num_steps,
                                          num_workers, worker_id)

    def real_batch(state):
        x, y, _w = next(data_it)
        return (torch.from_numpy(x).to(device),
                torch.from_numpy(y).to(device), state)

    prev_time, prev_step = time.time(), sess.global_step
    state = None
    for local_step in range(args.max_steps):
        batch = (real_batch(state) if data_it is not None else
                 lm1b.synthetic_batch(args.batch_size, args.num_steps,
                                      args.vocab_size, device, state))
        out = sess.run(batch)
        state = out.get('state')
        if local_step % args.log_frequency == 0 and worker_id == 0:
            now = time.time()
            num_words = args.batch_size * args.num_steps * num_workers
            wps = (out['global_step'] - prev_step) * num_words \
                / max(now - prev_time, 1e-9)
            parallax_amd.log.info(
                'Iteration %d, time = %.2fs, wps = %.0f, train loss = %.4f',
                out['global_step'], now - prev_time, wps,
                float(out['loss']))
            prev_time, prev_step = now, out['global_step']
    sess.close()


if __name__ == '__main__':
    main()
