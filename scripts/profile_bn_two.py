import os, sys, time
sys.path.insert(0, '/root/repo')
import torch
from parallax_amd.ops.batchnorm import ParallaxBatchNorm2d
dev='cuda:0'
torch.manual_seed(0)
x=torch.randn(64,64,112,112,device=dev).to(torch.bfloat16).to(memory_format=torch.channels_last)
dy=torch.randn_like(x)
m=ParallaxBatchNorm2d(64,momentum=0.001,relu=True).to(dev)
for _ in range(3):
    xx=x.detach().requires_grad_(True); m(xx).backward(dy)
torch.cuda.synchronize()
for _ in range(20):
    xx=x.detach().requires_grad_(True); m(xx).backward(dy)
torch.cuda.synchronize()
