"""Token embeddings (reference parity: contrib/text/embedding.py).

No network in this environment, so pretrained downloads are not
available; CustomEmbedding loads from a local file and TokenEmbedding
provides the indexing machinery (get_vecs_by_tokens / update_token_vectors).
"""
from __future__ import annotations

import numpy as np

from ... import ndarray as nd
from .vocab import Vocabulary


class TokenEmbedding(Vocabulary):
    def __init__(self, **kwargs):
        super().__init__(**kwargs)
        self._vec_len = 0
        self._idx_to_vec = None

    @property
    def vec_len(self):
        return self._vec_len

    @property
    def idx_to_vec(self):
        return self._idx_to_vec

    def _load_embedding_txt(self, file_path, elem_delim=" ",
                            encoding="utf8"):
        """Load `token v1 v2 ...` lines; index 0 (unknown) stays zero."""
        tokens, vecs = [], []
        with open(file_path, encoding=encoding) as f:
            for line in f:
                parts = line.rstrip().split(elem_delim)
                if len(parts) < 2:
                    continue
                tok, vals = parts[0], parts[1:]
                if self._vec_len == 0:
                    self._vec_len = len(vals)
                elif len(vals) != self._vec_len:
                    continue  # malformed line (reference skips + warns)
                tokens.append(tok)
                vecs.append([float(v) for v in vals])
        for t in tokens:
            if t not in self._token_to_idx:
                self._token_to_idx[t] = len(self._idx_to_token)
                self._idx_to_token.append(t)
        mat = np.zeros((len(self._idx_to_token), self._vec_len),
                       dtype=np.float32)
        for t, v in zip(tokens, vecs):
            mat[self._token_to_idx[t]] = v
        self._idx_to_vec = nd.array(mat)

    def get_vecs_by_tokens(self, tokens, lower_case_backup=False):
        single = isinstance(tokens, str)
        toks = [tokens] if single else tokens
        idx = []
        for t in toks:
            if t in self._token_to_idx:
                idx.append(self._token_to_idx[t])
            elif lower_case_backup and t.lower() in self._token_to_idx:
                idx.append(self._token_to_idx[t.lower()])
            else:
                idx.append(0)
        vecs = self._idx_to_vec[idx]
        return vecs[0] if single else vecs

    def update_token_vectors(self, tokens, new_vectors):
        toks = [tokens] if isinstance(tokens, str) else tokens
        arr = self._idx_to_vec.asnumpy()
        newv = new_vectors.asnumpy().reshape(len(toks), -1)
        for t, v in zip(toks, newv):
            if t not in self._token_to_idx:
                raise ValueError(f"token {t} unknown")
            arr[self._token_to_idx[t]] = v
        self._idx_to_vec = nd.array(arr)


class CustomEmbedding(TokenEmbedding):
    """Embedding loaded from a local `token v1 v2 ...` text file."""

    def __init__(self, pretrained_file_path, elem_delim=" ",
                 encoding="utf8", vocabulary=None, **kwargs):
        if vocabulary is not None:
            super().__init__(counter=None)
            self._token_to_idx = dict(vocabulary.token_to_idx)
            self._idx_to_token = list(vocabulary.idx_to_token)
        else:
            super().__init__(**kwargs)
        self._load_embedding_txt(pretrained_file_path, elem_delim,
                                 encoding)
