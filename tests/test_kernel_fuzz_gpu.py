"""Randomized shape fuzzing for the hand-written kernel family vs fp32
torch references (seeded, deterministic).  Guards the edge/tile-boundary
paths (clamped loads, guarded stores, masks, padding) that fixed-shape
unit tests can miss."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = 'cuda:0'


def _ext():
    from parallax_amd import ops
    if not torch.cuda.is_available():
        pytest.skip('needs GPU')
    if not ops.native_available():
        pytest.skip('native ext missing')
    return ops.ext()


def test_fuzz_gemm_family():
    e = _ext()
    import random
    rng = random.Random(123)
    for trial in range(14):
        M = rng.choice([1, 17, 100, 256, 300, 511, 513, 1000, 2048])
        N = rng.choice([16, 64, 100, 128, 192, 256, 500, 1024])
        K = 64 * rng.randint(1, 20)
        a = torch.rand(M, K, device=DEV, dtype=torch.bfloat16) * 2 - 1
        b = torch.rand(N, K, device=DEV, dtype=torch.bfloat16) * 2 - 1
        ref = a.float() @ b.float().t()
        den = float(ref.abs().max()) + 1e-6
        for name in ('pxgemm_bt', 'pxgemm8_bt', 'px_tall_bt'):
            if not hasattr(e, name):
                continue
            c = getattr(e, name)(a, b)
            rel = float((c.float() - ref).abs().max()) / den
            assert rel < 0.02, (name, M, N, K, rel)
        for v in (0, 1, 2, 3):
            c = e.pxgemm8_bt_v(a, b, v)
            rel = float((c.float() - ref).abs().max()) / den
            assert rel < 0.02, ('v%d' % v, M, N, K, rel)


def test_fuzz_skinny():
    e = _ext()
    if not hasattr(e, 'px_skinny_bt'):
        pytest.skip('no skinny')
    import random
    rng = random.Random(321)
    for trial in range(10):
        M = rng.randint(1, 128)
        N = 64 * rng.randint(1, 8)
        K = 64 * rng.randint(1, 40)
        a = torch.rand(M, K, device=DEV, dtype=torch.bfloat16) * 2 - 1
        b = torch.rand(N, K, device=DEV, dtype=torch.bfloat16) * 2 - 1
        ws = torch.zeros(M, N, dtype=torch.float32, device=DEV)
        cnt = torch.zeros(N // 64, dtype=torch.int32, device=DEV)
        out = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
        e.px_skinny_bt(a, b, ws, cnt, out, None)
        ref = a.float() @ b.float().t()
        den = float(ref.abs().max()) + 1e-6
        rel = float((out.float() - ref).abs().max()) / den
        assert rel < 0.02, (M, N, K, rel)
        assert int(cnt.abs().max()) == 0
        assert float(ws.abs().max()) == 0.0


def test_fuzz_attention_shapes():
    import torch.nn.functional as F
    from parallax_amd.ops.attention import px_attention, _eligible
    _ext()
    import random
    rng = random.Random(777)
    for trial in range(10):
        B = rng.randint(1, 6)
        H = rng.choice([1, 2, 5, 8])
        T = rng.randint(1, 64)
        S = rng.randint(1, 64)
        causal = rng.random() < 0.5 and T == S
        q = torch.randn(B, H, T, 64, device=DEV, dtype=torch.bfloat16)
        k = torch.randn(B, H, S, 64, device=DEV, dtype=torch.bfloat16)
        v = torch.randn(B, H, S, 64, device=DEV, dtype=torch.bfloat16)
        assert _eligible(q, k, v)
        q1 = q.clone().requires_grad_(True)
        k1 = k.clone().requires_grad_(True)
        v1 = v.clone().requires_grad_(True)
        o = px_attention(q1, k1, v1, is_causal=causal)
        ref = F.scaled_dot_product_attention(
            q.float(), k.float(), v.float(), is_causal=causal)
        assert torch.allclose(o.float(), ref, atol=5e-2), \
            (B, H, T, S, causal, (o.float() - ref).abs().max())
        do = torch.randn_like(o)
        o.backward(do)
        q2 = q.float().clone().requires_grad_(True)
        k2 = k.float().clone().requires_grad_(True)
        v2 = v.float().clone().requires_grad_(True)
        F.scaled_dot_product_attention(
            q2, k2, v2, is_causal=causal).backward(do.float())
        for g1, g2, nm in ((q1.grad, q2.grad, 'dq'),
                           (k1.grad, k2.grad, 'dk'),
                           (v1.grad, v2.grad, 'dv')):
            assert torch.allclose(g1.float(), g2, atol=8e-2), \
                (B, H, T, S, causal, nm,
                 (g1.float() - g2).abs().max())


def test_fuzz_pool_and_colsum():
    e = _ext()
    from parallax_amd.ops.pool import ParallaxMaxPool3x3s2
    import random
    rng = random.Random(55)
    for trial in range(6):
        N = rng.randint(1, 4)
        C = 8 * rng.randint(1, 12)
        H = rng.randint(3, 30)
        W = rng.randint(3, 30)
        x1 = (torch.randn(N, C, H, W, device=DEV).to(torch.bfloat16)
              .to(memory_format=torch.channels_last).requires_grad_(True))
        x2 = x1.detach().float().clone().requires_grad_(True)
        y1 = ParallaxMaxPool3x3s2()(x1)
        y2 = torch.nn.functional.max_pool2d(x2, 3, 2, 1)
        assert y1.shape == y2.shape, (N, C, H, W)
        assert torch.allclose(y1.float(), y2, atol=1e-2)
        dy = torch.randn_like(y2)
        y1.backward(dy.to(torch.bfloat16))
        y2.backward(dy)
        assert torch.allclose(x1.grad.float(), x2.grad, atol=6e-2), \
            (N, C, H, W)
    if hasattr(e, 'col_sum'):
        for trial in range(5):
            M = rng.randint(1, 5000)
            N = 8 * rng.randint(1, 256)
            x = torch.randn(M, N, device=DEV).to(torch.bfloat16)
            out = e.col_sum(x)
            ref = x.float().sum(0)
            assert torch.allclose(out, ref, atol=0.6, rtol=2e-3), (M, N)
