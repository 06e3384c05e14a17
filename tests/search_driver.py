"""Tiny driver used by test_partition_search.py: a partitioned-embedding
model trained long enough to cross the stat window, under the full
master/worker search harness."""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import parallax_amd
from parallax_amd import optim
from parallax_amd.ops.embedding import PartitionedEmbedding
from parallax_amd.session import TrainGraph


def build():
    torch.manual_seed(0)
    model = torch.nn.Module()
    model.emb = PartitionedEmbedding(64, 8, parallax_amd.get_partitioner(2))
    model.fc = torch.nn.Linear(8, 2)
    opt = optim.Adagrad(model.parameters(), lr=0.1)

    def step_fn(m, batch):
        ids, y = batch
        loss = ((m.fc(m.emb(ids)) - y) ** 2).mean()
        return loss, {}

    return TrainGraph(model, opt, step_fn)


def main():
    # master needs the minimum before spawning (reference: building the
    # single-GPU graph calls get_partitioner, runner.py:75-80)
    parallax_amd.get_partitioner(2)
    sess, num_workers, worker_id, _ = parallax_amd.parallel_run(
        build, sys.argv[1],
        parallax_config=parallax_amd.Config(run_option='HYBRID',
                                            search_partitions=True))
    print('worker %d running with %d partitions'
          % (worker_id, sess.graph.model.emb.num_partitions), flush=True)
    ids = torch.randint(0, 64, (16,))
    y = torch.randn(16, 2)
    for i in range(400):
        sess.run((ids, y))
    print('done p=%d' % sess.graph.model.emb.num_partitions,
          flush=True)
    sess.close()


if __name__ == '__main__':
    main()
