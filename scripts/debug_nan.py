"""Isolate the NaN: tiny LM1B config, eager vs graph, fused vs composed
sampled CE."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

os.environ.setdefault('WORLD_SIZE', '1')
os.environ.setdefault('RANK', '0')
os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
os.environ.setdefault('MASTER_PORT', '29561')

import parallax_amd
from parallax_amd.config import ParallaxConfig
from parallax_amd.models import lm1b as M


def run(tag, graph_mode, steps=30):
    import torch.distributed as dist
    os.environ['MASTER_PORT'] = str(29561 + (1 if graph_mode else 0))
    torch.manual_seed(0)
    sess, *_ = parallax_amd.parallel_run(
        lambda: M.build_lm1b_graph(batch_size=8, num_steps=4,
                                   vocab_size=5000, emb_dim=32,
                                   state_size=64, proj_size=32,
                                   num_sampled=128, min_partitions=4,
                                   lr=0.4),
        None, parallax_config=ParallaxConfig(run_option='HYBRID',
                                             search_partitions=False))
    x = torch.randint(0, 5000, (8, 4), device='cuda')
    y = torch.randint(0, 5000, (8, 4), device='cuda')
    if graph_mode:
        state0 = M.make_init_state(8, 64, 32, device='cuda',
                                   dtype=torch.bfloat16)
        sess.enable_cuda_graph((x, y, state0))
    losses = []
    first_nan = None
    for i in range(steps):
        out = sess.run((x, y, None))
        v = float(out['loss'])
        losses.append(v)
        if v != v and first_nan is None:
            first_nan = i
    print('%s: first_nan=%s losses[:6]=%s losses[-3:]=%s'
          % (tag, first_nan, ['%.3f' % v for v in losses[:6]],
             ['%.3f' % v for v in losses[-3:]]), flush=True)
    sess.close()
    if dist.is_initialized():
        dist.destroy_process_group()
    for k in ('WORLD_SIZE',):
        pass


if __name__ == '__main__':
    mode = sys.argv[1] if len(sys.argv) > 1 else 'both'
    if mode in ('eager', 'both'):
        run('eager', False)
    if mode in ('graph', 'both'):
        run('graph', True)
