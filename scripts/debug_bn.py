import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, torch.nn as nn
dev='cuda:0'
def t(tag, fn):
    try:
        fn(); torch.cuda.synchronize(); print(tag, 'OK', flush=True)
    except Exception as e:
        print(tag, 'EXC', repr(e), flush=True)
def bn(c, hw, b, cl=True, dt=torch.bfloat16, bwd=True):
    def fn():
        m = nn.BatchNorm2d(c).to(dev)
        x = torch.randn(b, c, hw, hw, device=dev, dtype=dt)
        if cl: x = x.to(memory_format=torch.channels_last)
        x.requires_grad_(True)
        z = m(x)
        if bwd: z.float().sum().backward()
    return fn
t('bn512@7 b2 bf16 cl fwdonly', bn(512,7,2,bwd=False))
t('bn512@7 b2 bf16 cl bwd', bn(512,7,2))
t('bn512@7 b2 bf16 NOcl bwd', bn(512,7,2,cl=False))
t('bn512@7 b2 fp32 cl bwd', bn(512,7,2,dt=torch.float32))
t('bn512@7 b4 bf16 cl bwd', bn(512,7,4))
t('bn512@7 b8 bf16 cl bwd', bn(512,7,8))
t('bn512@7 b16 bf16 cl bwd', bn(512,7,16))
t('bn512@7 b64 bf16 cl bwd', bn(512,7,64))
print('ALL DONE', flush=True)
