"""Tests for the NMT / skip_thoughts / CNN-family models (CPU, tiny
configs)."""

import pytest
import torch

import parallax_amd
from parallax_amd.config import ParallaxConfig
from parallax_amd.models import nmt, skip_thoughts, cnn
from parallax_amd.tagging import classify_params, split_dense_sparse


@pytest.fixture(autouse=True)
def _destroy_pg():
    yield
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()


def _worker_env(monkeypatch, port):
    monkeypatch.setenv('WORLD_SIZE', '1')
    monkeypatch.setenv('RANK', '0')
    monkeypatch.setenv('MASTER_ADDR', '127.0.0.1')
    monkeypatch.setenv('MASTER_PORT', str(port))


def test_nmt_transformer_tagging_and_step(monkeypatch, free_port):
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)
    sess, *_ = parallax_amd.parallel_run(
        lambda: nmt.build_nmt_graph(
            arch='transformer', src_vocab=500, tgt_vocab=500, d_model=32,
            heads=4, ffn=64, enc_layers=2, dec_layers=2, min_partitions=2,
            use_bf16=False),
        None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    assert sess.engine.mode == 'HYBRID'
    sparse, dense = split_dense_sparse(
        classify_params(sess.graph.model))
    names = {gi.name for gi in sparse}
    assert names == {'src_emb.weight', 'tgt_emb.weight'}
    assert len(dense) > 10
    batch = nmt.synthetic_batch(4, 6, 7, 500, 500)
    first = None
    for _ in range(8):
        out = sess.run(batch)
        if first is None:
            first = float(out['loss'])
    assert torch.isfinite(out['loss'])
    assert float(out['loss']) < first
    sess.close()


def test_nmt_pure_bf16_masters(monkeypatch, free_port):
    """pure_bf16: bf16 body + fp32 master weights in the optimizer, fp32
    sparse tables.  The eager (CPU) master path must learn and keep the
    bf16 param an exact rounded copy of its fp32 master."""
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)
    sess, *_ = parallax_amd.parallel_run(
        lambda: nmt.build_nmt_graph(
            arch='transformer', src_vocab=300, tgt_vocab=300, d_model=32,
            heads=4, ffn=64, enc_layers=1, dec_layers=1, min_partitions=2,
            use_bf16=False, pure_bf16=True, lr=3e-3),
        None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    m = sess.graph.model
    assert m.pos.dtype == torch.bfloat16
    assert m.src_emb.weight.dtype == torch.float32  # sparse plane stays
    batch = nmt.synthetic_batch(4, 5, 6, 300, 300)
    first = None
    for _ in range(10):
        out = sess.run(batch)
        if first is None:
            first = float(out['loss'])
    assert torch.isfinite(out['loss'])
    assert float(out['loss']) < first
    opt = sess.graph.optimizer
    mw = opt.state[m.pos].get('master')
    assert mw is not None and mw.dtype == torch.float32
    assert torch.equal(m.pos.detach(), mw.to(torch.bfloat16))
    # masters round-trip through the checkpoint payload uncast
    sd = opt.state_dict()
    opt.load_state_dict(sd)
    mw2 = opt.state[m.pos]['master']
    assert mw2.dtype == torch.float32 and torch.equal(mw2, mw)
    sess.close()


def test_gnmt_attention_step(monkeypatch, free_port):
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)
    sess, *_ = parallax_amd.parallel_run(
        lambda: nmt.build_nmt_graph(
            arch='gnmt', src_vocab=300, tgt_vocab=300, d_model=24,
            units=24, min_partitions=2, use_bf16=False),
        None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    batch = nmt.synthetic_batch(3, 5, 5, 300, 300)
    out = sess.run(batch)
    assert torch.isfinite(out['loss'])
    sess.close()


def test_skip_thoughts_step(monkeypatch, free_port):
    _worker_env(monkeypatch, free_port)
    torch.manual_seed(0)
    sess, *_ = parallax_amd.parallel_run(
        lambda: skip_thoughts.build_skip_thoughts_graph(
            vocab=400, emb_dim=16, units=24, num_sampled=32,
            min_partitions=2, use_bf16=False),
        None, parallax_config=ParallaxConfig(run_option='HYBRID'))
    batch = skip_thoughts.synthetic_batch(4, 6, 400)
    first = None
    for _ in range(6):
        out = sess.run(batch)
        if first is None:
            first = float(out['loss'])
    assert float(out['loss']) < first
    sess.close()


@pytest.mark.parametrize('name', ['vgg11', 'vgg16', 'vgg19', 'alexnet',
                                  'overfeat', 'googlenet', 'inception3',
                                  'inception4', 'lenet', 'trivial'])
def test_cnn_family_forward(name):
    torch.manual_seed(0)
    m = cnn.MODEL_REGISTRY[name](11)
    size = cnn.image_size_for(name)
    x = torch.randn(2, 3, size, size)
    out = m(x)
    assert out.shape == (2, 11)


def test_cnn_graph_optimizers():
    g = cnn.build_cnn_graph('trivial', num_classes=11,
                            optimizer_name='rmsprop', use_bf16=False,
                            channels_last=False)
    from parallax_amd import optim as pxo
    assert isinstance(g.optimizer, pxo.RMSProp)
    x = torch.randn(2, 3, 224, 224)
    y = torch.randint(0, 11, (2,))
    loss, _ = g.step_fn(g.model, (x, y))
    loss.backward()
    g.optimizer.step()
    assert torch.isfinite(loss)
    with pytest.raises(ValueError):
        cnn.build_cnn_graph('trivial', optimizer_name='adagrad')


def test_native_conv_state_dict_interchange():
    """Checkpoints move freely between the nn.Conv2d model and the
    native-conv swapped model (same parameter names/shapes)."""
    import os
    import torch
    from parallax_amd.models.resnet import ResNet
    torch.manual_seed(4)
    ref = ResNet(depth=50, num_classes=10)
    os.environ['PARALLAX_NATIVE_CONV1X1'] = '1'
    os.environ['PARALLAX_NATIVE_CONV3X3'] = '1'
    try:
        from parallax_amd.ops.conv import swap_conv1x1, swap_conv3x3
        import copy
        swapped = swap_conv3x3(swap_conv1x1(copy.deepcopy(ref)))
        n_swapped = sum(
            1 for m in swapped.modules()
            if type(m).__name__ in ('ParallaxConv1x1', 'ParallaxConv3x3'))
        assert n_swapped > 30, n_swapped
        sd = ref.state_dict()
        swapped.load_state_dict(sd)          # nn -> native
        ref2 = ResNet(depth=50, num_classes=10)
        ref2.load_state_dict(swapped.state_dict())  # native -> nn
        for k, v in ref2.state_dict().items():
            assert torch.equal(v, sd[k]), k
    finally:
        del os.environ['PARALLAX_NATIVE_CONV1X1']
        del os.environ['PARALLAX_NATIVE_CONV3X3']


def test_resnet_v2_step():
    """Pre-activation ResNet v2 (reference resnet_model.py:129-180)
    builds and trains a step on CPU."""
    import torch
    from parallax_amd.models.resnet import ResNet
    torch.manual_seed(5)
    m = ResNet(depth=50, num_classes=10, version=2)
    x = torch.randn(2, 3, 64, 64)
    y = torch.randint(0, 10, (2,))
    logits = m(x)
    loss = torch.nn.functional.cross_entropy(logits, y)
    loss.backward()
    assert torch.isfinite(loss)
    grads = [p.grad for p in m.parameters() if p.grad is not None]
    assert len(grads) > 100


def test_cnn_extra_families():
    """model_config.py:29-63 parity: vgg11/lenet/inception4 + the
    cifar10 registry build and produce correct logit shapes."""
    import torch
    from parallax_amd.models.cnn import get_model, image_size_for
    from parallax_amd.models.cnn_extra import MODEL_REGISTRY_CIFAR
    torch.manual_seed(6)
    for name in ('vgg11', 'lenet'):
        s = image_size_for(name)
        y = get_model(name, 1001)(torch.randn(1, 3, s, s))
        assert y.shape == (1, 1001), name
    assert image_size_for('inception4') == 299
    for name in ('resnet20', 'resnet110_v2', 'densenet40_k12', 'alexnet',
                 'trivial'):
        assert name in MODEL_REGISTRY_CIFAR
        m = get_model(name, 11, dataset='cifar10')
        y = m(torch.randn(2, 3, 32, 32))
        assert y.shape == (2, 11), name
    # dataset-keyed lookup: cifar 'alexnet' is the tutorial net, not the
    # imagenet one
    assert type(get_model('alexnet', 11, 'cifar10')).__name__ \
        == 'AlexNetCifar'


def test_inception4_share_columns():
    """The 'share' split-branch columns in inception_v4_c reuse the
    previous column's stem (inception_model.py:25-35): the shared stem
    conv runs once and feeds both heads."""
    import torch
    from parallax_amd.models.cnn_extra import _inc4_c
    m = _inc4_c(1536)
    x = torch.randn(1, 1536, 4, 4)
    y = m(x)
    assert y.shape == (1, 1536, 4, 4)
    # column 3 shares col 2's first op; columns 5 shares col 4's 3-op stem
    assert m.n_share == [0, 0, 0, 1, 0, 3]
    y.sum().backward()


def test_cnn_graph_cifar_step():
    from parallax_amd.models.cnn import build_cnn_graph
    import torch
    torch.manual_seed(7)
    g = build_cnn_graph('resnet20', num_classes=11, batch_size=4,
                        use_bf16=False, dataset='cifar10')
    x = torch.randn(4, 3, 32, 32)
    y = torch.randint(0, 11, (4,))
    loss, _ = g.step_fn(g.model, (x, y))
    loss.backward()
    assert torch.isfinite(loss)


def test_lr_schedules_reference_values():
    """resnet_model.py:232-255 warmup+piecewise; cifar schedules
    :326-331 / densenet_model.py:96-103; tf exponential_decay."""
    from parallax_amd import lr_schedule as L
    bpe = L.IMAGENET_NUM_TRAIN_IMAGES / 64.0
    # warmup: 0.1 at step 0 (batch 64 -> scaled lr stays 0.1, so flat)
    assert L.resnet_imagenet_lr(0, 64) == pytest.approx(0.1)
    assert L.resnet_imagenet_lr(int(bpe * 4), 64) == pytest.approx(0.1)
    # decades at epochs 30/60/80
    assert L.resnet_imagenet_lr(int(bpe * 31), 64) == pytest.approx(0.01)
    assert L.resnet_imagenet_lr(int(bpe * 61), 64) == pytest.approx(1e-3)
    assert L.resnet_imagenet_lr(int(bpe * 81), 64) == pytest.approx(1e-4)
    # global batch > 256 scales the post-warmup LR linearly
    big = L.resnet_imagenet_lr(10 ** 9, 64, num_workers=8)  # global 512
    assert big == pytest.approx(512 / 256.0 * 0.1 / 10 ** 3)
    # warmup is linear from 0.1 to the scaled LR
    mid = L.resnet_imagenet_lr(
        int(L.IMAGENET_NUM_TRAIN_IMAGES / 512.0 * 2.5), 64, 8)
    assert 0.1 < mid < 0.2
    # cifar resnet
    cb = int(L.CIFAR10_NUM_TRAIN_IMAGES / 128)
    assert L.resnet_cifar_lr(cb * 82, 128) == pytest.approx(0.1)
    assert L.resnet_cifar_lr(cb * 83, 128) == pytest.approx(0.01)
    assert L.resnet_cifar_lr(cb * 500, 128) == pytest.approx(0.0002)
    # densenet cifar
    assert L.densenet_cifar_lr(cb * 151, 128) == pytest.approx(0.01)
    # exponential decay (staircase)
    assert L.exponential_decay(0.1, 99, 100, 0.5) == pytest.approx(0.1)
    assert L.exponential_decay(0.1, 250, 100, 0.5) == \
        pytest.approx(0.025)
    # dispatch
    assert L.model_lr_schedule('resnet50') is L.resnet_imagenet_lr
    assert L.model_lr_schedule('resnet20', 'cifar10') is L.resnet_cifar_lr
    assert L.model_lr_schedule('densenet40_k12', 'cifar10') \
        is L.densenet_cifar_lr
    assert L.model_lr_schedule('vgg16') is None
    # set_lr
    import torch as T
    from parallax_amd import optim
    opt = optim.SGD([T.nn.Parameter(T.zeros(2))], lr=0.5)
    L.set_lr(opt, 0.125)
    assert opt.param_groups[0]['lr'] == 0.125


def test_topk_accuracy_values():
    import torch as T
    from parallax_amd.models.cnn import topk_accuracy
    logits = T.tensor([[9., 1., 2., 3., 4., 5., 0.],
                       [0., 9., 8., 7., 6., 5., 1.]])
    acc = topk_accuracy(logits, T.tensor([0, 0]))
    assert float(acc['top_1']) == 0.5        # row0 right, row1 wrong
    assert float(acc['top_5']) == 0.5        # label 0 not in row1 top-5
    acc = topk_accuracy(logits, T.tensor([5, 5]))
    assert float(acc['top_1']) == 0.0
    assert float(acc['top_5']) == 1.0


def test_topk_accuracy_clamps_to_classes():
    import torch as T
    from parallax_amd.models.cnn import topk_accuracy
    logits = T.tensor([[1.0, 2.0, 3.0]])          # 3 classes < k=5
    acc = topk_accuracy(logits, T.tensor([0]))
    assert float(acc['top_1']) == 0.0
    assert float(acc['top_5']) == 1.0             # k clamped to 3
