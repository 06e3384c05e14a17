"""LM1B LSTM language model — the flagship hybrid (sparse+dense) workload.

Reference config (examples/lm1b/language_model.py:16-100,
language_model_graph.py:16-65): vocab 793,470; embedding 512 partitioned
into >=32 shards via get_partitioner; 1-layer LSTM state 2048 projected to
512; sampled softmax with 8192 candidates over softmax_w [V,512] /
softmax_b [V]; Adagrad(0.2) with clip_by_global_norm(10) on the dense
(LSTM) params; batch 128 x 20 unrolled steps.

Sparse parameters (classified by the tagging pass, updated through the
in-process PS): the embedding shards, softmax_w shards and softmax_b.
Dense parameters (RCCL all-reduce): LSTM W/B/W_proj.
"""

import torch
import torch.nn as nn

from parallax_amd import optim
from parallax_amd.ops.embedding import PartitionedEmbedding
from parallax_amd.ops.lstm import ParallaxLSTM
from parallax_amd.ops.sampled_softmax import sampled_softmax_loss
from parallax_amd.ops.misc import clip_by_global_norm
from parallax_amd.partitions import get_partitioner
from parallax_amd.session import TrainGraph

VOCAB_SIZE = 793470
EMB_DIM = 512
STATE_SIZE = 2048
PROJ_SIZE = 512
NUM_STEPS = 20
BATCH_SIZE = 128
NUM_SAMPLED = 8192


class LM1BModel(nn.Module):
    def __init__(self, vocab_size=VOCAB_SIZE, emb_dim=EMB_DIM,
                 state_size=STATE_SIZE, proj_size=PROJ_SIZE,
                 num_sampled=NUM_SAMPLED, min_partitions=32,
                 dtype=torch.float32):
        super().__init__()
        self.vocab_size = vocab_size
        self.num_sampled = num_sampled
        partitioner = get_partitioner(min_partitions)
        self.emb = PartitionedEmbedding(vocab_size, emb_dim, partitioner,
                                        dtype=dtype, init_scale=0.1)
        self.lstm = ParallaxLSTM(emb_dim, state_size, proj_size,
                                 dtype=dtype)
        self.softmax_w = PartitionedEmbedding(vocab_size, proj_size,
                                              partitioner, dtype=dtype,
                                              init_scale=proj_size ** -0.5)
        self.softmax_b = nn.Parameter(torch.zeros(vocab_size, dtype=dtype))
        from parallax_amd.tagging import mark_sparse
        mark_sparse(self.softmax_b)

    # reference keep_prob = 0.9 during training (language_model.py:24):
    # dropout on the embedded inputs (:69) and on each timestep's
    # projected output before the softmax (:87 — the RECURRENT h stays
    # un-dropped, so one dropout over the stacked outputs is equivalent)
    dropout_p = 0.1

    def forward(self, x, y, state=None):
        """x, y: [B, T] int64.  Returns (mean loss per word, final state)."""
        emb = self.emb(x)                      # [B, T, D]
        body = self.lstm.w.dtype               # bf16 in pure-bf16 mode
        if emb.dtype != body and not torch.is_autocast_enabled():
            emb = emb.to(body)
        if self.training and self.dropout_p > 0:
            emb = torch.nn.functional.dropout(emb, self.dropout_p)
        hs, state = self.lstm(emb, state)      # [B, T, P]
        if self.training and self.dropout_p > 0:
            hs = torch.nn.functional.dropout(hs, self.dropout_p)
        flat_h = hs.reshape(-1, hs.shape[-1])
        flat_y = y.reshape(-1)
        losses = sampled_softmax_loss(self.softmax_w, self.softmax_b,
                                      flat_y, flat_h, self.num_sampled,
                                      self.vocab_size)
        return losses.mean(), state

    def dense_parameters(self):
        return list(self.lstm.parameters())


def build_lm1b_graph(batch_size=BATCH_SIZE, num_steps=NUM_STEPS,
                     vocab_size=VOCAB_SIZE, lr=0.2, min_partitions=32,
                     emb_dim=EMB_DIM, state_size=STATE_SIZE,
                     proj_size=PROJ_SIZE, num_sampled=NUM_SAMPLED,
                     max_grad_norm=10.0, dtype=torch.float32,
                     use_bf16=True, ema_decay=0.999, pure_bf16=False):
    """use_bf16: run the GEMM path under bf16 autocast on GPU (fp32 master
    weights, fp32 CE and optimizer math).
    pure_bf16: the dense LSTM body lives in bf16 (no per-step autocast
    weight casts); fp32 update precision comes from the optimizers'
    fused master-weight path; the sparse tables and EMA shadows stay
    fp32.
    ema_decay: exponential moving average of the LSTM params (reference
    language_model_graph.py:63-65, decay .999); shadows live as model
    buffers (-> broadcast + checkpointed); None disables."""
    model = LM1BModel(vocab_size, emb_dim, state_size, proj_size,
                      num_sampled, min_partitions, dtype=dtype)
    if pure_bf16:
        model.lstm.to(torch.bfloat16)
    optimizer = optim.Adagrad(model.parameters(), lr=lr,
                              initial_accumulator_value=1.0)
    dense_params = model.dense_parameters()

    def step_fn(m, batch):
        x, y, state = batch
        if use_bf16 and not pure_bf16 and x.is_cuda:
            with torch.autocast('cuda', dtype=torch.bfloat16):
                loss, new_state = m(x, y, state)
        else:
            loss, new_state = m(x, y, state)
        return loss, {'state': (new_state[0].detach(),
                                new_state[1].detach())}

    graph = TrainGraph(model, optimizer, step_fn)

    # clip_by_global_norm on the dense (LSTM) grads, applied between
    # backward and the optimizer step (reference
    # language_model_graph.py:53).  Installed as an engine pre-step hook by
    # wrapping the optimizer step.
    n_ema = 0
    if ema_decay is not None:
        for i, p in enumerate(dense_params):
            # fp32 shadows even over a bf16 body: .999-decay drift would
            # be swallowed by bf16 rounding
            model.register_buffer('ema_%d' % i,
                                  p.detach().clone().float())
        n_ema = len(dense_params)

    orig_step = optimizer.step

    def step_with_clip(closure=None):
        grads = [p.grad for p in dense_params if p.grad is not None
                 and p.grad.layout == torch.strided]
        if grads:
            clip_by_global_norm(grads, max_grad_norm)
        out = orig_step(closure)
        if n_ema:
            from parallax_amd.ops.misc import ema_update
            shadows = [getattr(model, 'ema_%d' % i) for i in range(n_ema)]
            ema_update(shadows, [p.detach() for p in dense_params],
                       ema_decay)
        return out

    optimizer.step = step_with_clip
    return graph


def make_init_state(batch_size=BATCH_SIZE, state_size=STATE_SIZE,
                    proj_size=PROJ_SIZE, device='cpu',
                    dtype=torch.float32):
    """Zero LSTM state (h [B, P], c [B, H]) — explicit tensors so the
    hipGraph capture can feed the recurrent state back in-graph."""
    h = torch.zeros(batch_size, proj_size, device=device, dtype=dtype)
    c = torch.zeros(batch_size, state_size, device=device, dtype=dtype)
    return (h, c)


def synthetic_batch(batch_size=BATCH_SIZE, num_steps=NUM_STEPS,
                    vocab_size=VOCAB_SIZE, device='cpu', state=None,
                    generator=None):
    """Synthetic LM1B data (reference use_synthetic flag,
    lm1b_distributed_driver.py:81-84: uniform random ids)."""
    x = torch.randint(0, vocab_size, (batch_size, num_steps), device=device,
                      generator=generator)
    y = torch.randint(0, vocab_size, (batch_size, num_steps), device=device,
                      generator=generator)
    return (x, y, state)
