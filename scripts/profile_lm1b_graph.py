"""Graph-mode LM1B step breakdown: replay vs sparse-engine vs optimizer."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

os.environ.setdefault('WORLD_SIZE', '1')
os.environ.setdefault('RANK', '0')
os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
os.environ.setdefault('MASTER_PORT', '29532')

import parallax_amd
from parallax_amd.config import ParallaxConfig
from parallax_amd.models import lm1b as M


def main():
    steps = int(os.environ.get('PROF_STEPS', '20'))
    sess, *_ = parallax_amd.parallel_run(
        lambda: M.build_lm1b_graph(),
        None, parallax_config=ParallaxConfig(run_option='HYBRID',
                                             search_partitions=False))
    dev = sess.comm.device
    state0 = M.make_init_state(device=dev, dtype=torch.bfloat16)
    sess.enable_cuda_graph(M.synthetic_batch(device=dev, state=state0))
    g = sess.graph
    eng = sess.engine

    def sync():
        torch.cuda.synchronize()

    for _ in range(5):
        sess.run(M.synthetic_batch(device=dev))

    acc = {'copy+replay': 0.0, 'sparse': 0.0, 'opt': 0.0, 'total': 0.0}
    for _ in range(steps):
        batch = M.synthetic_batch(device=dev)
        sync()
        T0 = time.perf_counter()
        for dst, src in zip(sess._static_batch, batch):
            if isinstance(dst, torch.Tensor) and isinstance(src, torch.Tensor):
                dst.copy_(src, non_blocking=True)
        sess._graph_obj.replay()
        sync()
        t1 = time.perf_counter()
        if eng.sparse_engine:
            eng.sparse_engine.reduce_and_apply()
        sync()
        t2 = time.perf_counter()
        if eng.dense_engine:
            eng.dense_engine.reduce_gradients()
        g.optimizer.step()
        sync()
        t3 = time.perf_counter()
        acc['copy+replay'] += t1 - T0
        acc['sparse'] += t2 - t1
        acc['opt'] += t3 - t2
        acc['total'] += t3 - T0
    print('=== graphed LM1B breakdown (ms/step) ===')
    for k, v in acc.items():
        print('%12s: %8.3f' % (k, v / steps * 1000))
    sess.close()


if __name__ == '__main__':
    main()
