"""Minimal parallax_amd driver: 2->1 linear regression, GradientDescent.

The MI355X-native counterpart of the reference's
examples/simple/simple_driver.py: the same single-GPU definition runs
data-parallel through ``parallel_run``; the master process spawns one
worker per GPU (or one CPU worker) and each worker gets back a session.

Run: ``python simple_driver.py [--learning_rate 0.01] [--steps 200]``
"""

import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import parallax_amd
from parallax_amd.models.simple import build_simple_graph

train_x = np.array([
    [3.3, 1.7], [4.4, 2.7], [5.5, 3.6], [6.71, 4.0], [6.93, 4.1],
    [4.168, 2.5], [9.779, 5.0], [6.182, 3.1], [7.59, 4.2], [2.167, 1.2],
], dtype=np.float32)
train_y = (train_x @ np.array([[1.3], [-0.5]], dtype=np.float32) + 0.7)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--learning_rate', type=float, default=0.01)
    ap.add_argument('--steps', type=int, default=200)
    ap.add_argument('--resource_info', default=os.path.join(
        os.path.dirname(os.path.abspath(__file__)), 'resource_info'))
    args = ap.parse_args()

    sess, num_workers, worker_id, num_replicas = parallax_amd.parallel_run(
        lambda: build_simple_graph(lr=args.learning_rate),
        args.resource_info,
        parallax_config=parallax_amd.Config(run_option='HYBRID',
                                            search_partitions=False))
    device = sess.comm.device
    xs = torch.from_numpy(train_x).to(device)
    ys = torch.from_numpy(train_y).to(device)
    n = xs.shape[0]
    cursor = worker_id  # round-robin per worker (reference feed pattern)
    for i in range(args.steps):
        j = cursor % n
        cursor += num_workers
        out = sess.run((xs[j:j + 1], ys[j:j + 1]))
        if i % 20 == 0 and worker_id == 0:
            print('global step: %d, loss: %f'
                  % (out['global_step'], float(out['loss'])))
    if worker_id == 0:
        print('final loss: %f' % float(out['loss']))
    sess.close()


if __name__ == '__main__':
    main()
