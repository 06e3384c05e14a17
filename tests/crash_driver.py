"""Driver whose worker fails mid-training — used by test_failstop.py."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import parallax_amd
from parallax_amd.models.simple import build_simple_graph


def main():
    sess, num_workers, worker_id, _ = parallax_amd.parallel_run(
        build_simple_graph, sys.argv[1],
        parallax_config=parallax_amd.Config(search_partitions=False))
    x = torch.randn(4, 2)
    y = torch.randn(4, 1)
    for i in range(100):
        sess.run((x, y))
        if i == 5 and worker_id == num_workers - 1:
            # only the LAST worker dies: at world>1 the others keep
            # looping until the master's fail-stop kills the group
            raise RuntimeError('injected worker failure')


if __name__ == '__main__':
    main()
