"""Section-level timing of one LM1B training step (GPU): where do the
milliseconds go?  fwd / bwd / sparse-engine / dense+optimizer, with
torch.cuda.synchronize between sections (sync overhead itself is measured
by the 'sync-only' row)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

os.environ.setdefault('WORLD_SIZE', '1')
os.environ.setdefault('RANK', '0')
os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
os.environ.setdefault('MASTER_PORT', '29531')

import parallax_amd
from parallax_amd.config import ParallaxConfig
from parallax_amd.models import lm1b as M


def main():
    steps = int(os.environ.get('PROF_STEPS', '10'))
    sess, *_ = parallax_amd.parallel_run(
        lambda: M.build_lm1b_graph(),
        None, parallax_config=ParallaxConfig(run_option='HYBRID',
                                             search_partitions=False))
    dev = sess.comm.device
    g = sess.graph
    eng = sess.engine

    def sync():
        torch.cuda.synchronize()

    # warmup
    state = None
    for _ in range(5):
        out = sess.run(M.synthetic_batch(device=dev, state=state))
        state = out.get('state')

    acc = {'data': 0.0, 'fwd': 0.0, 'bwd': 0.0, 'sparse': 0.0,
           'dense+opt': 0.0, 'total': 0.0}
    for _ in range(steps):
        sync()
        T0 = time.perf_counter()
        batch = M.synthetic_batch(device=dev, state=state)
        sync()
        t1 = time.perf_counter()
        eng.prepare_step()
        g.optimizer.zero_grad(set_to_none=True)
        loss, fetches = g.step_fn(g.model, batch)
        state = (fetches['state'][0], fetches['state'][1])
        sync()
        t2 = time.perf_counter()
        loss.backward()
        sync()
        t3 = time.perf_counter()
        if eng.dense_engine:
            eng.dense_engine.reduce_gradients()
        if eng.sparse_engine:
            eng.sparse_engine.reduce_and_apply()
        sync()
        t4 = time.perf_counter()
        g.optimizer.step()
        sync()
        t5 = time.perf_counter()
        acc['data'] += t1 - T0
        acc['fwd'] += t2 - t1
        acc['bwd'] += t3 - t2
        acc['sparse'] += t4 - t3
        acc['dense+opt'] += t5 - t4
        acc['total'] += t5 - T0

    print('=== LM1B step breakdown (%d steps, ms/step) ===' % steps)
    for k, v in acc.items():
        print('%12s: %8.3f' % (k, v / steps * 1000))

    # host-vs-device: time the same loop without syncs
    sync()
    t0 = time.perf_counter()
    for _ in range(steps):
        out = sess.run(M.synthetic_batch(device=dev, state=state))
        state = out.get('state')
    sync()
    t1 = time.perf_counter()
    print('  nosync run: %8.3f ms/step' % ((t1 - t0) / steps * 1000))
    sess.close()


if __name__ == '__main__':
    main()
