import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from parallax_amd.ops.attention import px_attention, _eligible
from parallax_amd import ops
e = ops.ext()
B, H, T, D = 4, 8, 48, 64
base = torch.randn(3, B, T, H, D)
q, k, v = [base[i].to('cuda').to(torch.bfloat16).transpose(1, 2)
           .requires_grad_(True) for i in range(3)]
print('strides', q.stride(), 'eligible', _eligible(q, k, v), flush=True)
for causal in (False, True):
    o = px_attention(q, k, v, is_causal=causal)
    torch.cuda.synchronize()
    print('fwd ok causal=%s' % causal, flush=True)
    refs = [base[i].float().transpose(1, 2).requires_grad_(True)
            for i in range(3)]
    o2 = torch.nn.functional.scaled_dot_product_attention(
        *refs, is_causal=causal)
    err = (o.detach().cpu().float() - o2.detach()).abs().max()
    print('fwd maxerr', float(err), flush=True)
    do = torch.randn(B, H, T, D)
    o.backward(do.to('cuda').to(torch.bfloat16))
    torch.cuda.synchronize()
    print('bwd ok', flush=True)
    o2.backward(do)
    for got, ref, name in ((q, refs[0], 'dq'), (k, refs[1], 'dk'),
                           (v, refs[2], 'dv')):
        print(name, float((got.grad.cpu().float() - ref.grad).abs().max()),
              flush=True)
        got.grad = None
print('cross shape', flush=True)
q = torch.randn(2, 8, 40, 64, device='cuda', dtype=torch.bfloat16,
                requires_grad=True)
k = torch.randn(2, 8, 56, 64, device='cuda', dtype=torch.bfloat16,
                requires_grad=True)
v = torch.randn(2, 8, 56, 64, device='cuda', dtype=torch.bfloat16,
                requires_grad=True)
o = px_attention(q, k, v)
o.sum().backward()
torch.cuda.synchronize()
print('cross ok', flush=True)
