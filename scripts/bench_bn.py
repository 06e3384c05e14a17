"""Isolated fwd+bwd A/B of the fused NHWC BN vs nn.BatchNorm2d (+relu
[+add]) on the ResNet-50 batch-64 shapes, within-probe interleaved."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from parallax_amd.ops.batchnorm import ParallaxBatchNorm2d

dev='cuda:0'
torch.manual_seed(0)

def run(N,C,H,W,add=False,iters=30):
    x=torch.randn(N,C,H,W,device=dev).to(torch.bfloat16).to(memory_format=torch.channels_last)
    idt=torch.randn_like(x) if add else None
    dy=torch.randn_like(x)
    ours=ParallaxBatchNorm2d(C,momentum=0.001,relu=True).to(dev)
    ref=torch.nn.BatchNorm2d(C,momentum=0.001).to(dev)
    def f_ours():
        xx=x.detach().requires_grad_(True)
        y=ours(xx,addend=idt) if add else ours(xx)
        y.backward(dy)
    def f_ref():
        xx=x.detach().requires_grad_(True)
        ii=idt.detach().requires_grad_(True) if add else None
        y=ref(xx)
        y=torch.relu(y+ii) if add else torch.relu(y)
        y.backward(dy)
    out={}
    for name,fn in (('ours',f_ours),('miopen',f_ref)):
        for _ in range(5): fn()
        torch.cuda.synchronize(); t0=time.perf_counter()
        for _ in range(iters): fn()
        torch.cuda.synchronize()
        out[name]=(time.perf_counter()-t0)/iters*1e6
    mb=N*C*H*W*2/1e6
    print('N%d C%4d %3dx%-3d add=%d  ours %8.1f us  miopen %8.1f us'
          '  (%.0f MB)' % (N, C, H, W, add, out['ours'], out['miopen'],
                           mb), flush=True)

if __name__=='__main__':
    for (C,HW) in ((64,112),(256,56),(64,56),(512,28),(128,28),(1024,14),(256,14),(2048,7),(512,7)):
        run(64,C,HW,HW)
    for (C,HW) in ((256,56),(512,28),(1024,14),(2048,7)):
        run(64,C,HW,HW,add=True)
