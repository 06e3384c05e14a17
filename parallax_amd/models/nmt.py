"""NMT models — hybrid PS+AR workload with partitioned src/tgt embeddings.

Reference: examples/nmt (~8k LoC GNMT/attention seq2seq: LSTM enc-dec with
Luong attention, BPE vocabs, ``parallax.shard.shard`` input pipeline and
``get_partitioner`` embeddings — model_helper.py:310-311,
iterator_utils.py:103).  BASELINE.json config 4 names the Transformer
variant ("NMT Transformer hybrid PS+AR, sparse src/tgt embeddings
partitioned"); both are provided:

  * NMTTransformer — encoder-decoder Transformer; src/tgt embeddings are
    PartitionedEmbedding tables (sparse/PS plane), projection + attention
    + FFN params dense (RCCL all-reduce plane); LayerNorm is the
    hand-written HIP kernel; loss is the fused softmax-CE over the shared
    output embedding (tied weights, indexed-slices grads).
  * GNMTAttention — the reference's architecture: LSTM encoder + LSTM
    decoder with Luong-style multiplicative attention, fused gate kernels.
"""

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from parallax_amd import optim
from parallax_amd.ops.embedding import PartitionedEmbedding
from parallax_amd.ops.layernorm import ParallaxLayerNorm
from parallax_amd.ops.linear import PxLinear
from parallax_amd.ops.lstm import ParallaxLSTM
from parallax_amd.ops.sampled_softmax import sampled_softmax_loss
from parallax_amd.ops.softmax_ce import softmax_cross_entropy
from parallax_amd.partitions import get_partitioner
from parallax_amd.session import TrainGraph

SRC_VOCAB = 32000
TGT_VOCAB = 32000


class _MHA(nn.Module):
    """Self- (fused qkv) or cross- (separate q / kv) attention.  Only the
    projections the instance actually runs are created — an unused dense
    parameter would produce no gradient and trip the dense engine's
    incomplete-bucket guard at world_size > 1."""

    def __init__(self, d, heads, cross=False):
        super().__init__()
        self.d = d
        self.h = heads
        if cross:
            self.kv = PxLinear(d, 2 * d, bias=False)
            self.q = PxLinear(d, d, bias=False)
        else:
            self.qkv = PxLinear(d, 3 * d, bias=False)
        self.out = PxLinear(d, d, bias=False)

    def forward(self, x, mem=None, causal=False):
        B, T, D = x.shape
        if mem is None:
            q, k, v = self.qkv(x).chunk(3, dim=-1)
        else:
            q = self.q(x)
            k, v = self.kv(mem).chunk(2, dim=-1)
        S = k.shape[1]

        def split(t, n):
            return t.view(B, n, self.h, D // self.h).transpose(1, 2)

        q, k, v = split(q, T), split(k, S), split(v, S)
        from parallax_amd.ops.attention import px_attention
        o = px_attention(q, k, v, is_causal=causal)
        return self.out(o.transpose(1, 2).reshape(B, T, D))


class _Block(nn.Module):
    def __init__(self, d, heads, ffn, cross=False, dropout=0.1):
        super().__init__()
        self.p = dropout
        self.ln1 = ParallaxLayerNorm(d)
        self.attn = _MHA(d, heads)
        self.cross = cross
        if cross:
            self.ln_x = ParallaxLayerNorm(d)
            self.xattn = _MHA(d, heads, cross=True)
        self.ln2 = ParallaxLayerNorm(d)
        self.fc1 = PxLinear(d, ffn)   # K=512: pxgemm wins
        self.fc2 = nn.Linear(ffn, d)  # K=2048: hipBLASLt wins

    def _drop(self, t):
        if self.training and self.p > 0:
            return F.dropout(t, self.p)
        return t

    def forward(self, x, mem=None, causal=False):
        # residual dropout (the reference NMT's dropout=0.2-class
        # regularization; standard transformer placement)
        x = x + self._drop(self.attn(self.ln1(x), causal=causal))
        if self.cross:
            x = x + self._drop(self.xattn(self.ln_x(x), mem=mem))
        x = x + self._drop(self.fc2(F.gelu(self.fc1(self.ln2(x)))))
        return x


class NMTTransformer(nn.Module):
    def __init__(self, src_vocab=SRC_VOCAB, tgt_vocab=TGT_VOCAB, d_model=512,
                 heads=8, ffn=2048, enc_layers=6, dec_layers=6,
                 min_partitions=8, max_len=512):
        super().__init__()
        part = get_partitioner(min_partitions)
        self.d_model = d_model
        self.tgt_vocab = tgt_vocab
        self.src_emb = PartitionedEmbedding(src_vocab, d_model, part,
                                            init_scale=0.05)
        self.tgt_emb = PartitionedEmbedding(tgt_vocab, d_model, part,
                                            init_scale=0.05)
        self.pos = nn.Parameter(torch.zeros(max_len, d_model))
        from parallax_amd.tagging import mark_dense
        mark_dense(self.pos)
        self.enc = nn.ModuleList(
            [_Block(d_model, heads, ffn) for _ in range(enc_layers)])
        self.dec = nn.ModuleList(
            [_Block(d_model, heads, ffn, cross=True)
             for _ in range(dec_layers)])
        self.ln_f = ParallaxLayerNorm(d_model)

    def forward(self, src, tgt_in, tgt_out):
        scale = math.sqrt(self.d_model)
        # lookups come from the (always-fp32) sparse tables; cast to the
        # body dtype (bf16 in pure-bf16 mode, no-op otherwise)
        body = self.pos.dtype
        e = self.src_emb(src).to(body) * scale + self.pos[:src.shape[1]]
        for blk in self.enc:
            e = blk(e)
        d = self.tgt_emb(tgt_in).to(body) * scale \
            + self.pos[:tgt_in.shape[1]]
        for blk in self.dec:
            d = blk(d, mem=e, causal=True)
        d = self.ln_f(d)
        # Tied output softmax over the partitioned tgt table, via sampled
        # softmax so the table's gradient stays indexed-slices (a full
        # dense logits GEMM would produce a dense [V, D] grad and push the
        # whole table onto the all-reduce plane).
        flat_d = d.reshape(-1, self.d_model)
        flat_y = tgt_out.reshape(-1)
        losses = sampled_softmax_loss(self.tgt_emb, None, flat_y, flat_d,
                                      num_sampled=min(4096,
                                                      self.tgt_vocab // 2),
                                      num_classes=self.tgt_vocab)
        return losses.mean()

    @torch.no_grad()
    def encode(self, src):
        scale = math.sqrt(self.d_model)
        body = self.pos.dtype
        e = self.src_emb(src).to(body) * scale + self.pos[:src.shape[1]]
        for blk in self.enc:
            e = blk(e)
        return e

    @torch.no_grad()
    def greedy_decode(self, src, sos_id=1, eos_id=2, max_len=64):
        """Inference-time greedy decoding over the FULL tied softmax
        (sampled softmax is train-only; eval scores every class —
        reference nmt/model.py infer path / evaluation_utils BLEU).
        Returns a list of id-lists (without SOS, truncated at EOS)."""
        mem = self.encode(src)
        B = src.shape[0]
        W = self.tgt_emb.full_weight().to(mem.dtype)   # [V, D]
        ys = torch.full((B, 1), sos_id, dtype=torch.long,
                        device=src.device)
        done = torch.zeros(B, dtype=torch.bool, device=src.device)
        scale = math.sqrt(self.d_model)
        for _ in range(max_len):
            d = self.tgt_emb(ys).to(mem.dtype) * scale \
                + self.pos[:ys.shape[1]]
            for blk in self.dec:
                d = blk(d, mem=mem, causal=True)
            d = self.ln_f(d)[:, -1]                    # [B, D]
            nxt = (d @ W.t()).argmax(dim=-1)           # [B]
            nxt = torch.where(done, torch.full_like(nxt, eos_id), nxt)
            ys = torch.cat([ys, nxt[:, None]], dim=1)
            done |= nxt == eos_id
            if bool(done.all()):
                break
        outs = []
        for row in ys[:, 1:].tolist():
            hyp = []
            for t in row:
                if t == eos_id:
                    break
                hyp.append(t)
            outs.append(hyp)
        return outs

    @torch.no_grad()
    def beam_decode(self, src, beam_width=4, length_penalty_weight=0.0,
                    sos_id=1, eos_id=2, max_len=64):
        """Beam-search decoding with the GNMT length penalty
        lp(l) = ((5 + l) / 6) ** alpha; candidates are ranked by
        cumulative log-prob / lp at every step (reference infer path:
        nmt/model.py:425-439 BeamSearchDecoder with beam_width and
        length_penalty_weight).  Returns the best hypothesis per
        sentence (id-lists without SOS/EOS); beam_width=1 with alpha=0
        reduces to greedy_decode."""
        mem = self.encode(src)
        B, S, D = mem.shape
        K = beam_width
        dev = src.device
        W = self.tgt_emb.full_weight().to(mem.dtype)    # [V, D]
        V = W.shape[0]
        scale = math.sqrt(self.d_model)
        mem = mem.unsqueeze(1).expand(B, K, S, D).reshape(B * K, S, D)
        ys = torch.full((B * K, 1), sos_id, dtype=torch.long, device=dev)
        # only beam 0 is live at the start (all beams hold identical
        # prefixes; duplicates would waste the beam)
        scores = torch.full((B, K), float('-inf'), device=dev)
        scores[:, 0] = 0.0
        done = torch.zeros(B * K, dtype=torch.bool, device=dev)
        # true hypothesis length per beam (finished beams stop growing —
        # their EOS padding is free AND must not change their penalty)
        lens = torch.zeros(B * K, dtype=torch.float32, device=dev)

        def lp(length):
            if length_penalty_weight == 0.0:
                return torch.ones_like(length)
            return ((5.0 + length) / 6.0) ** length_penalty_weight

        for _ in range(max_len):
            d = self.tgt_emb(ys).to(mem.dtype) * scale \
                + self.pos[:ys.shape[1]]
            for blk in self.dec:
                d = blk(d, mem=mem, causal=True)
            d = self.ln_f(d)[:, -1]                     # [B*K, D]
            logp = torch.log_softmax((d @ W.t()).float(), dim=-1)
            # finished beams: the only continuation is EOS at no cost
            logp = torch.where(
                done[:, None],
                torch.full_like(logp, float('-inf')).scatter(
                    1, torch.full((B * K, 1), eos_id, dtype=torch.long,
                                  device=dev), 0.0),
                logp)
            cum = scores.reshape(B * K, 1) + logp       # [B*K, V]
            new_len = torch.where(done, lens, lens + 1)  # [B*K]
            ranked = cum / lp(new_len)[:, None]
            top = ranked.reshape(B, K * V).topk(K, dim=-1).indices
            beam_idx = top // V                         # [B, K]
            tok = top % V                               # [B, K]
            flat_src = (torch.arange(B, device=dev)[:, None] * K +
                        beam_idx).reshape(-1)
            scores = cum.reshape(B, K * V).gather(1, top)
            ys = torch.cat([ys[flat_src], tok.reshape(-1, 1)], dim=1)
            lens = new_len[flat_src]
            done = done[flat_src] | (tok.reshape(-1) == eos_id)
            if bool(done.all()):
                break

        final = (scores / lp(lens).reshape(B, K)).argmax(dim=1)  # [B]
        outs = []
        rows = ys[:, 1:].reshape(B, K, -1)
        for b in range(B):
            hyp = []
            for t in rows[b, final[b]].tolist():
                if t == eos_id:
                    break
                hyp.append(t)
            outs.append(hyp)
        return outs


class GNMTAttention(nn.Module):
    """Reference-architecture seq2seq: LSTM encoder, LSTM decoder with
    Luong multiplicative attention (nmt/model.py attention_architecture)."""

    def __init__(self, src_vocab=SRC_VOCAB, tgt_vocab=TGT_VOCAB,
                 emb_dim=512, units=512, min_partitions=8):
        super().__init__()
        part = get_partitioner(min_partitions)
        self.src_emb = PartitionedEmbedding(src_vocab, emb_dim, part,
                                            init_scale=0.05)
        self.tgt_emb = PartitionedEmbedding(tgt_vocab, emb_dim, part,
                                            init_scale=0.05)
        self.encoder = ParallaxLSTM(emb_dim, units)
        self.decoder = ParallaxLSTM(emb_dim + units, units)
        self.att_w = nn.Parameter(torch.empty(units, units).uniform_(-0.05,
                                                                     0.05))
        self.proj = nn.Linear(2 * units, units)
        self.units = units
        self.tgt_vocab = tgt_vocab

    dropout_p = 0.2   # reference nmt hparams dropout default

    def _dec_step(self, emb_t, ctx, dec_h, dec_c, enc_out):
        """One decoder step: LSTM cell + Luong multiplicative attention.
        Returns (attended output, new ctx, new h, new c)."""
        x = torch.cat([emb_t, ctx], dim=-1)
        dec_h, dec_c = self.decoder.step(x, dec_h, dec_c)
        scores = torch.einsum('bu,uv,bsv->bs', dec_h, self.att_w, enc_out)
        alpha = torch.softmax(scores, dim=-1)
        ctx = torch.einsum('bs,bsu->bu', alpha, enc_out)
        out = torch.tanh(self.proj(torch.cat([dec_h, ctx], dim=-1)))
        return out, ctx, dec_h, dec_c

    def forward(self, src, tgt_in, tgt_out):
        semb = self.src_emb(src)
        if self.training and self.dropout_p > 0:
            semb = torch.nn.functional.dropout(semb, self.dropout_p)
        enc_out, (h, c) = self.encoder(semb)
        B, S, U = enc_out.shape
        T = tgt_in.shape[1]
        emb = self.tgt_emb(tgt_in)
        ctx = enc_out.new_zeros(B, U)
        dec_h, dec_c = h, c
        outs = []
        for t in range(T):
            o, ctx, dec_h, dec_c = self._dec_step(emb[:, t], ctx, dec_h,
                                                  dec_c, enc_out)
            outs.append(o)
        dec = torch.stack(outs, dim=1).reshape(B * T, U)
        losses = sampled_softmax_loss(self.tgt_emb, None,
                                      tgt_out.reshape(-1), dec,
                                      num_sampled=min(1024,
                                                      self.tgt_vocab // 2),
                                      num_classes=self.tgt_vocab)
        return losses.mean()

    @torch.no_grad()
    def greedy_decode(self, src, sos_id=1, eos_id=2, max_len=64):
        """Step-wise greedy inference over the full tied softmax
        (reference gnmt infer path; same output convention as
        NMTTransformer.greedy_decode)."""
        enc_out, (h, c) = self.encoder(self.src_emb(src))
        B, S, U = enc_out.shape
        W = self.tgt_emb.full_weight()                  # [V, D]
        ctx = enc_out.new_zeros(B, U)
        ys = torch.full((B,), sos_id, dtype=torch.long, device=src.device)
        done = torch.zeros(B, dtype=torch.bool, device=src.device)
        rows = []
        for _ in range(max_len):
            o, ctx, h, c = self._dec_step(self.tgt_emb(ys), ctx, h, c,
                                          enc_out)
            nxt = (o @ W.t()).argmax(dim=-1)
            nxt = torch.where(done, torch.full_like(nxt, eos_id), nxt)
            rows.append(nxt)
            done |= nxt == eos_id
            ys = nxt
            if bool(done.all()):
                break
        outs = []
        for row in torch.stack(rows, dim=1).tolist():
            hyp = []
            for t in row:
                if t == eos_id:
                    break
                hyp.append(t)
            outs.append(hyp)
        return outs

    @torch.no_grad()
    def beam_decode(self, src, beam_width=10, length_penalty_weight=0.0,
                    sos_id=1, eos_id=2, max_len=64):
        """Beam search with the GNMT length penalty (the reference's GNMT
        infer defaults: beam 10; same candidate-ranking convention as
        NMTTransformer.beam_decode — cumulative log-prob / lp at every
        step, finished beams pad with free EOS)."""
        enc_out, (h, c) = self.encoder(self.src_emb(src))
        B, S, U = enc_out.shape
        K = beam_width
        dev = src.device
        W = self.tgt_emb.full_weight()
        V = W.shape[0]
        rep = lambda t: (t.unsqueeze(1).expand(B, K, *t.shape[1:])
                         .reshape(B * K, *t.shape[1:]))
        enc_out = rep(enc_out)
        h, c = rep(h).clone(), rep(c).clone()
        ctx = enc_out.new_zeros(B * K, U)
        cur = torch.full((B * K,), sos_id, dtype=torch.long, device=dev)
        seqs = torch.zeros(B * K, 0, dtype=torch.long, device=dev)
        scores = torch.full((B, K), float('-inf'), device=dev)
        scores[:, 0] = 0.0
        done = torch.zeros(B * K, dtype=torch.bool, device=dev)
        lens = torch.zeros(B * K, dtype=torch.float32, device=dev)

        def lp(length):
            if length_penalty_weight == 0.0:
                return torch.ones_like(length)
            return ((5.0 + length) / 6.0) ** length_penalty_weight

        for _ in range(max_len):
            o, ctx, h, c = self._dec_step(self.tgt_emb(cur), ctx, h, c,
                                          enc_out)
            logp = torch.log_softmax((o @ W.t()).float(), dim=-1)
            logp = torch.where(
                done[:, None],
                torch.full_like(logp, float('-inf')).scatter(
                    1, torch.full((B * K, 1), eos_id, dtype=torch.long,
                                  device=dev), 0.0),
                logp)
            cum = scores.reshape(B * K, 1) + logp
            new_len = torch.where(done, lens, lens + 1)
            ranked = cum / lp(new_len)[:, None]
            top = ranked.reshape(B, K * V).topk(K, dim=-1).indices
            beam_idx = top // V
            tok = top % V
            flat_src = (torch.arange(B, device=dev)[:, None] * K +
                        beam_idx).reshape(-1)
            scores = cum.reshape(B, K * V).gather(1, top)
            seqs = torch.cat([seqs[flat_src], tok.reshape(-1, 1)], dim=1)
            ctx, h, c = ctx[flat_src], h[flat_src], c[flat_src]
            lens = new_len[flat_src]
            done = done[flat_src] | (tok.reshape(-1) == eos_id)
            cur = tok.reshape(-1)
            if bool(done.all()):
                break

        final = (scores / lp(lens).reshape(B, K)).argmax(dim=1)
        rows = seqs.reshape(B, K, -1)
        outs = []
        for b in range(B):
            hyp = []
            for t in rows[b, final[b]].tolist():
                if t == eos_id:
                    break
                hyp.append(t)
            outs.append(hyp)
        return outs


def build_nmt_graph(arch='transformer', src_vocab=SRC_VOCAB,
                    tgt_vocab=TGT_VOCAB, d_model=512, heads=8, ffn=2048,
                    enc_layers=6, dec_layers=6, min_partitions=8, lr=1e-3,
                    use_bf16=True, pure_bf16=False, **kw):
    """pure_bf16 (transformer only): the dense body lives in bf16 — no
    per-step autocast weight-cast kernels — with fp32 update precision via
    the optimizers' fused master-weight path; the sparse embedding tables
    stay fp32 (row updates keep full precision)."""
    if arch == 'transformer':
        model = NMTTransformer(src_vocab, tgt_vocab, d_model, heads, ffn,
                               enc_layers, dec_layers, min_partitions)
        if pure_bf16:
            model.to(torch.bfloat16)
            model.src_emb.float()
            model.tgt_emb.float()
    else:
        pure_bf16 = False
        model = GNMTAttention(src_vocab, tgt_vocab, d_model,
                              kw.get('units', d_model), min_partitions)
    optimizer = optim.Adam(model.parameters(), lr=lr)

    def step_fn(m, batch):
        src, tgt_in, tgt_out = batch
        if use_bf16 and not pure_bf16 and src.is_cuda:
            with torch.autocast('cuda', dtype=torch.bfloat16):
                loss = m(src, tgt_in, tgt_out)
        else:
            loss = m(src, tgt_in, tgt_out)
        return loss, {}

    return TrainGraph(model, optimizer, step_fn)


def synthetic_batch(batch_size=64, src_len=48, tgt_len=48,
                    src_vocab=SRC_VOCAB, tgt_vocab=TGT_VOCAB, device='cpu'):
    src = torch.randint(0, src_vocab, (batch_size, src_len), device=device)
    tgt_in = torch.randint(0, tgt_vocab, (batch_size, tgt_len),
                           device=device)
    tgt_out = torch.randint(0, tgt_vocab, (batch_size, tgt_len),
                            device=device)
    return src, tgt_in, tgt_out
