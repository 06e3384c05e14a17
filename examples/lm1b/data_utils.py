"""LM1B data utilities — real-corpus path.

Counterpart of the reference's examples/lm1b/data_utils.py: a vocabulary
over a token file and a dataset that iterates (x, y, weight) batches with
round-robin assignment of files/sentences to workers
(reference data_utils.py:133-140: sentence i goes to worker
i % num_workers == worker_id).
"""

import numpy as np


class Vocabulary(object):
    """Token -> id mapping from a vocab file (one token per line; <S>, </S>
    and <UNK> expected, reference data_utils.py Vocabulary)."""

    def __init__(self, path):
        self._token_to_id = {}
        self._tokens = []
        with open(path) as f:
            for line in f:
                tok = line.strip()
                if tok and tok not in self._token_to_id:
                    self._token_to_id[tok] = len(self._tokens)
                    self._tokens.append(tok)
        self.bos = self._token_to_id.get('<S>', 0)
        self.eos = self._token_to_id.get('</S>', 0)
        self.unk = self._token_to_id.get('<UNK>', 0)

    @property
    def size(self):
        return len(self._tokens)

    def token_to_id(self, tok):
        return self._token_to_id.get(tok, self.unk)

    def id_to_token(self, i):
        return self._tokens[i]

    def encode(self, sentence):
        return ([self.bos] +
                [self.token_to_id(t) for t in sentence.split()] +
                [self.eos])


class LM1BDataset(object):
    """Sentence corpus -> (x, y, w) LM batches.

    ``iterate_forever(batch_size, num_steps, num_workers, worker_id)``
    yields [B, T] int arrays where each batch row is an independent
    sentence stream; sentences are assigned round-robin to workers
    (reference data_utils.py:133-140)."""

    def __init__(self, corpus_path, vocab):
        self.vocab = vocab
        with open(corpus_path) as f:
            self.sentences = [line.strip() for line in f if line.strip()]

    def _sentence_stream(self, num_workers, worker_id):
        i = 0
        while True:
            if i % num_workers == worker_id:
                yield self.vocab.encode(self.sentences[i %
                                                       len(self.sentences)])
            i += 1

    def iterate_forever(self, batch_size, num_steps, num_workers=1,
                        worker_id=0):
        stream = self._sentence_stream(num_workers, worker_id)
        buffers = [[] for _ in range(batch_size)]
        while True:
            x = np.zeros((batch_size, num_steps), dtype=np.int64)
            y = np.zeros((batch_size, num_steps), dtype=np.int64)
            w = np.zeros((batch_size, num_steps), dtype=np.float32)
            for b in range(batch_size):
                while len(buffers[b]) < num_steps + 1:
                    buffers[b].extend(next(stream))
                ids = buffers[b][:num_steps + 1]
                buffers[b] = buffers[b][num_steps:]
                x[b] = ids[:-1]
                y[b] = ids[1:]
                w[b] = 1.0
            yield x, y, w
