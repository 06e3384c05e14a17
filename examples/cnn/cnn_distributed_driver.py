"""CNN benchmark driver — tf_cnn_benchmarks counterpart.

Reference: examples/tf_cnn_benchmarks/CNNBenchmark_distributed_driver.py
(+ benchmark_cnn.py): model registry, synthetic ImageNet, images/sec
logging, momentum/sgd/rmsprop.
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import parallax_amd
from parallax_amd.models import cnn, resnet


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--model', default='resnet50')
    ap.add_argument('--dataset', default='imagenet',
                    choices=['imagenet', 'cifar10'])
    ap.add_argument('--batch_size', type=int, default=64)
    ap.add_argument('--optimizer', default='momentum',
                    choices=['momentum', 'sgd', 'rmsprop'])
    ap.add_argument('--learning_rate', type=float, default=None,
                    help='fixed LR; default: the model\'s published '
                         'schedule (warmup+piecewise for resnet, '
                         'piecewise for cifar resnet/densenet) or 0.1')
    ap.add_argument('--max_steps', type=int, default=1000000)
    ap.add_argument('--log_frequency', type=int, default=10)
    ap.add_argument('--run_option', default='MPI')
    ap.add_argument('--params_stat', action='store_true')
    ap.add_argument('--data_dir', default=None,
                    help='image-folder dataset root (class subdirs); '
                         'default: synthetic images')
    ap.add_argument('--distortions', action='store_true')
    ap.add_argument('--resource_info', default=os.path.join(
        os.path.dirname(os.path.abspath(__file__)), 'resource_info'))
    args = ap.parse_args()

    from parallax_amd import lr_schedule
    schedule = (None if args.learning_rate is not None
                else lr_schedule.model_lr_schedule(args.model,
                                                   args.dataset))
    lr0 = args.learning_rate if args.learning_rate is not None else 0.1

    sess, num_workers, worker_id, _ = parallax_amd.parallel_run(
        lambda: cnn.build_cnn_graph(args.model,
                                    num_classes=(11 if args.dataset ==
                                                 'cifar10' else 1001),
                                    batch_size=args.batch_size,
                                    optimizer_name=args.optimizer,
                                    lr=lr0,
                                    dataset=args.dataset),
        args.resource_info,
        parallax_config=parallax_amd.Config(run_option=args.run_option))
    device = sess.comm.device
    if args.params_stat and worker_id == 0:
        stats = sess.params_stat()
        parallax_amd.log.info('params: %d tensors, %.1f M elements, '
                              '%.1f MB grads/step', stats['num_tensors'],
                              stats['num_elements'] / 1e6,
                              stats['grad_bytes'] / 1e6)
    data_it = None
    if args.data_dir:
        import preprocessing
        from parallax_amd.data import PrefetchLoader
        data_it = iter(PrefetchLoader(
            preprocessing.minibatch_iterator(
                args.data_dir, args.batch_size,
                cnn.image_size_for(args.model, args.dataset),
                train=True, distortions=args.distortions,
                seed=worker_id),
            device))
    prev = time.time()
    for step in range(args.max_steps):
        if schedule is not None:
            lr_schedule.set_lr(
                sess.graph.optimizer,
                schedule(step, args.batch_size, num_workers))
        if data_it is not None:
            x, y = next(data_it)
        else:
            x, y = resnet.synthetic_batch(
                args.batch_size,
                image_size=cnn.image_size_for(args.model, args.dataset),
                num_classes=11 if args.dataset == 'cifar10' else 1001,
                device=device)
        out = sess.run((x, y))
        if step % args.log_frequency == 0 and worker_id == 0:
            now = time.time()
            ips = args.log_frequency * args.batch_size * num_workers \
                / max(now - prev, 1e-9)
            parallax_amd.log.info(
                'step %d, images/sec = %.1f, loss = %.4f',
                out['global_step'], ips, float(out['loss']))
            prev = now
    sess.close()


if __name__ == '__main__':
    main()
