"""CNN benchmark driver — tf_cnn_benchmarks counterpart.

Reference: examples/tf_cnn_benchmarks/CNNBenchmark_distributed_driver.py
(+ benchmark_cnn.py): model registry, synthetic ImageNet or real images,
images/sec logging, momentum/sgd/rmsprop, per-model LR schedules,
optional training accuracy, checkpointing, and an eval mode
(benchmark_cnn.py:588-640 _eval_once top-1/top-5).
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import parallax_amd
import parallax_config
from parallax_amd.models import cnn, resnet


def parse_args():
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
    ap.add_argument('--print_training_accuracy', action='store_true',
                    help='fetch top-1/top-5 of each logged batch '
                         '(benchmark_cnn print_training_accuracy)')
    ap.add_argument('--data_dir', default=None,
                    help='image-folder dataset root (class subdirs); '
                         'default: synthetic images')
    ap.add_argument('--distortions', action='store_true')
    ap.add_argument('--ckpt_dir', default=None)
    ap.add_argument('--save_ckpt_steps', type=int, default=None)
    ap.add_argument('--save_ckpt_secs', type=int, default=None)
    ap.add_argument('--eval', action='store_true',
                    help='evaluate the latest --ckpt_dir checkpoint '
                         '(top-1/top-5) instead of training')
    ap.add_argument('--num_eval_batches', type=int, default=50)
    ap.add_argument('--resource_info', default=os.path.join(
        os.path.dirname(os.path.abspath(__file__)), 'resource_info'))
    return ap.parse_args()


def num_classes_for(dataset):
    return 11 if dataset == 'cifar10' else 1001


def evaluate(args):
    """Single-process eval pass (reference _eval_once): build the model,
    restore the latest checkpoint, report mean top-1/top-5."""
    torch.manual_seed(0)
    nc = num_classes_for(args.dataset)
    model = cnn.get_model(args.model, nc, args.dataset).eval()
    if args.ckpt_dir:
        from parallax_amd.checkpoint import latest_checkpoint
        payload = torch.load(latest_checkpoint(args.ckpt_dir),
                             map_location='cpu', weights_only=True)
        model.load_state_dict(payload['model'])
    size = cnn.image_size_for(args.model, args.dataset)
    if args.data_dir:
        import preprocessing
        batches = preprocessing.minibatch_iterator(
            args.data_dir, args.batch_size, size, train=False,
            num_shards=1, shard_id=0, epochs=1)
    else:
        batches = (resnet.synthetic_batch(args.batch_size, image_size=size,
                                          num_classes=nc)
                   for _ in range(args.num_eval_batches))
    t1 = t5 = n = 0.0
    with torch.no_grad():
        for i, (x, y) in enumerate(batches):
            if i >= args.num_eval_batches:
                break
            acc = cnn.topk_accuracy(model(x).float(), y)
            t1 += float(acc['top_1'])
            t5 += float(acc['top_5'])
            n += 1
    print('eval: batches=%d top_1=%.4f top_5=%.4f' %
          (int(n), t1 / max(n, 1), t5 / max(n, 1)))


def train(args):
    from parallax_amd import lr_schedule
    schedule = (None if args.learning_rate is not None
                else lr_schedule.model_lr_schedule(args.model,
                                                   args.dataset))
    lr0 = args.learning_rate if args.learning_rate is not None else 0.1
    nc = num_classes_for(args.dataset)

    sess, num_workers, worker_id, _ = parallax_amd.parallel_run(
        lambda: cnn.build_cnn_graph(
            args.model, num_classes=nc, batch_size=args.batch_size,
            optimizer_name=args.optimizer, lr=lr0, dataset=args.dataset,
            with_accuracy=args.print_training_accuracy),
        args.resource_info,
        parallax_config=parallax_config.build_config(args))
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
                num_classes=nc, device=device)
        out = sess.run((x, y))
        if step % args.log_frequency == 0 and worker_id == 0:
            now = time.time()
            ips = args.log_frequency * args.batch_size * num_workers \
                / max(now - prev, 1e-9)
            extra = ''
            if args.print_training_accuracy:
                extra = ', top_1 = %.3f, top_5 = %.3f' % (
                    float(out['top_1']), float(out['top_5']))
            parallax_amd.log.info(
                'step %d, images/sec = %.1f, loss = %.4f%s',
                out['global_step'], ips, float(out['loss']), extra)
            prev = now
    sess.close()


def main():
    args = parse_args()
    if args.eval:
        evaluate(args)
    else:
        train(args)


if __name__ == '__main__':
    main()
