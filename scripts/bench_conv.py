"""Hand-written NHWC conv kernels vs MIOpen (torch conv2d), fwd-only."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from parallax_amd.ops.conv import conv3x3_nhwc, conv1x1_nhwc
torch.manual_seed(0)
torch.backends.cudnn.benchmark = True
dev = 'cuda:0'

def t(fn, iters=30):
    for _ in range(8): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e6

print('--- conv3x3 fwd (bf16 channels_last, batch 64) ---', flush=True)
for (C, HW, COUT, S) in ((64, 56, 64, 1), (128, 28, 128, 1),
                         (256, 14, 256, 1), (512, 7, 512, 1),
                         (128, 56, 128, 2), (256, 28, 256, 2)):
    x = torch.randn(64, C, HW, HW, device=dev, dtype=torch.bfloat16).to(
        memory_format=torch.channels_last)
    w = torch.randn(COUT, C, 3, 3, device=dev, dtype=torch.bfloat16)*0.1
    wcl = w.to(memory_format=torch.channels_last)
    with torch.no_grad():
        t_ours = t(lambda: conv3x3_nhwc(x, w, stride=S))
        t_mi = t(lambda: torch.nn.functional.conv2d(x, wcl, stride=S,
                                                    padding=1))
    fl = 2.0*64*(HW//S)*(HW//S)*COUT*9*C
    print('3x3 C=%4d HW=%2d COUT=%4d s=%d  ours %7.1f us (%6.1f TF)  '
          'miopen %7.1f us (%6.1f TF)' % (C, HW, COUT, S, t_ours,
          fl/t_ours/1e6, t_mi, fl/t_mi/1e6), flush=True)

print('--- conv1x1 fwd ---', flush=True)
for (C, HW, COUT) in ((64, 56, 256), (256, 56, 64), (512, 28, 128),
                      (1024, 14, 256), (2048, 7, 512)):
    x = torch.randn(64, C, HW, HW, device=dev, dtype=torch.bfloat16).to(
        memory_format=torch.channels_last)
    w = torch.randn(COUT, C, 1, 1, device=dev, dtype=torch.bfloat16)*0.1
    wcl = w.to(memory_format=torch.channels_last)
    with torch.no_grad():
        t_ours = t(lambda: conv1x1_nhwc(x, w))
        t_mi = t(lambda: torch.nn.functional.conv2d(x, wcl))
    fl = 2.0*64*HW*HW*COUT*C
    print('1x1 C=%4d HW=%2d COUT=%4d      ours %7.1f us (%6.1f TF)  '
          'miopen %7.1f us (%6.1f TF)' % (C, HW, COUT, t_ours,
          fl/t_ours/1e6, t_mi, fl/t_mi/1e6), flush=True)
