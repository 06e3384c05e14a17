"""Skip-thoughts inference layer: sentence -> thought-vector encoding.

Reference: examples/skip_thoughts/skip_thoughts_encoder.py (pad/batch,
tokenize, optional expanded-embedding feed, L2-normalised output) and
encoder_manager.py (load several trained models, concatenate their
encodings).  Checkpoints are the framework's own chief-written
``model.ckpt-N.pt`` files (parallax_amd/checkpoint.py layout).
"""

import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

from parallax_amd.models.skip_thoughts import SkipThoughts  # noqa: E402

EOS = '</s>'


def _tokenize(item):
    if isinstance(item, str):
        return item.split()
    return list(item)


class SkipThoughtsEncoder(object):
    """Encodes sentences with one trained skip-thoughts model.

    vocab maps word -> id for the trained embedding table; unknown words
    fall back to id 0 unless an expanded ``embedding_dict``
    (vocabulary_expansion output, word -> vector) is supplied, in which
    case tokens are fed as vectors and any word in the expanded
    vocabulary is usable (skip_thoughts_encoder.py:205-227)."""

    def __init__(self, model, vocab, embedding_dict=None):
        self.model = model.eval()
        self.vocab = vocab
        self.embedding_dict = embedding_dict
        self.emb_dim = model.emb.embedding_dim

    @classmethod
    def from_checkpoint(cls, ckpt_dir, vocab_map, embedding_dict=None,
                        **model_kwargs):
        from parallax_amd.checkpoint import latest_checkpoint
        path = latest_checkpoint(ckpt_dir)
        if path is None:
            raise FileNotFoundError('no checkpoint under %s' % ckpt_dir)
        payload = torch.load(path, map_location='cpu', weights_only=True)
        model = SkipThoughts(**model_kwargs)
        model.load_state_dict(payload['model'])
        return cls(model, vocab_map, embedding_dict)

    def _embed_sentence(self, words):
        if self.embedding_dict is None:
            ids = torch.tensor([self.vocab.get(w, 0) for w in words],
                               dtype=torch.long)
            with torch.no_grad():
                return self.model.emb(ids)
        vecs = [self.embedding_dict[w] for w in words
                if w in self.embedding_dict]
        if not vecs:
            return torch.zeros(1, self.emb_dim)
        return torch.tensor(np.stack(vecs), dtype=torch.float32)

    def encode(self, data, use_norm=True, batch_size=128, use_eos=False):
        """data: list of sentences (strings or token lists).  Returns a
        list of thought vectors (numpy, one per sentence)."""
        embedded = []
        for item in data:
            words = _tokenize(item)
            if use_eos:
                words = words + [EOS]
            if not words:
                words = [EOS]
            embedded.append(self._embed_sentence(words))
        out = []
        for lo in range(0, len(embedded), batch_size):
            chunk = embedded[lo:lo + batch_size]
            L = max(e.shape[0] for e in chunk)
            xs = torch.zeros(len(chunk), L, self.emb_dim)
            for i, e in enumerate(chunk):
                xs[i, :e.shape[0]] = e
            with torch.no_grad():
                # right-zero padding: run full length, take each
                # sentence's true final state
                hs, _ = self.model.encoder(xs)
            for i, e in enumerate(chunk):
                v = hs[i, e.shape[0] - 1].numpy()
                if use_norm:
                    v = v / max(np.linalg.norm(v), 1e-12)
                out.append(v)
        return out


class EncoderManager(object):
    """encoder_manager.py equivalent: multiple models, concatenated
    encodings."""

    def __init__(self):
        self.encoders = []

    def load_model(self, ckpt_dir, vocab_map, embedding_dict=None,
                   **model_kwargs):
        self.encoders.append(SkipThoughtsEncoder.from_checkpoint(
            ckpt_dir, vocab_map, embedding_dict, **model_kwargs))

    def add_encoder(self, encoder):
        self.encoders.append(encoder)

    def encode(self, data, **kwargs):
        if not self.encoders:
            raise ValueError('Must call load_model at least once '
                             'before calling encode')
        encoded = [e.encode(data, **kwargs) for e in self.encoders]
        return [np.concatenate(vs) for vs in zip(*encoded)]
