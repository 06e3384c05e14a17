"""Throughput smoke of the tf_cnn_benchmarks model family (reference
models beyond ResNet: VGG, AlexNet, GoogLeNet, InceptionV3, Overfeat,
Trivial) — one timed training loop each, bf16 autocast, batch 64."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault('WORLD_SIZE', '1'); os.environ.setdefault('RANK', '0')
os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
os.environ.setdefault('MASTER_PORT', '29571')
import torch
import parallax_amd
from parallax_amd.config import ParallaxConfig
from parallax_amd.models import cnn as M
from parallax_amd.models import resnet as R

names = ['vgg16', 'vgg19', 'alexnet', 'googlenet', 'inception3',
         'overfeat', 'trivial', 'resnet101', 'resnet152']
for name in names:
    try:
        os.environ['MASTER_PORT'] = str(29571)
        import torch.distributed as dist
        if dist.is_initialized():
            pass
        sess, _, _, _ = parallax_amd.parallel_run(
            lambda: M.build_cnn_graph(model_name=name, batch_size=64),
            None, parallax_config=ParallaxConfig(run_option='MPI'))
        dev = sess.comm.device
        isz = M.image_size_for(name)
        b = R.synthetic_batch(64, image_size=isz, device=dev)
        for _ in range(8):
            sess.run(b)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(30):
            sess.run(b)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 30
        print('%-12s %8.1f img/s  (%.2f ms/step)'
              % (name, 64 / dt, dt * 1e3), flush=True)
        sess.close()
    except Exception as e:
        print('%-12s FAILED: %r' % (name, e), flush=True)
