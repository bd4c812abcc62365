"""Vocabulary (reference parity: contrib/text/vocab.py)."""
from __future__ import annotations


class Vocabulary:
    """Indexes tokens by frequency; reserves index 0 for unknown.

    API parity: token_to_idx / idx_to_token / unknown_token /
    reserved_tokens / to_indices / to_tokens / len().
    """

    def __init__(self, counter=None, most_freq_count=None, min_freq=1,
                 unknown_token="<unk>", reserved_tokens=None):
        assert min_freq > 0
        self._unknown_token = unknown_token
        self._reserved_tokens = list(reserved_tokens) if reserved_tokens \
            else None
        self._idx_to_token = [unknown_token]
        if reserved_tokens:
            assert unknown_token not in reserved_tokens
            assert len(set(reserved_tokens)) == len(reserved_tokens)
            self._idx_to_token.extend(reserved_tokens)
        self._token_to_idx = {t: i for i, t in
                              enumerate(self._idx_to_token)}
        if counter is not None:
            self._index_counter_keys(counter, most_freq_count, min_freq)

    def _index_counter_keys(self, counter, most_freq_count, min_freq):
        pairs = sorted(counter.items(), key=lambda kv: (-kv[1], kv[0]))
        taken = 0
        for token, freq in pairs:
            if freq < min_freq:
                break
            if most_freq_count is not None and taken >= most_freq_count:
                break
            if token in self._token_to_idx:
                continue
            self._token_to_idx[token] = len(self._idx_to_token)
            self._idx_to_token.append(token)
            taken += 1

    def __len__(self):
        return len(self._idx_to_token)

    @property
    def token_to_idx(self):
        return self._token_to_idx

    @property
    def idx_to_token(self):
        return self._idx_to_token

    @property
    def unknown_token(self):
        return self._unknown_token

    @property
    def reserved_tokens(self):
        return self._reserved_tokens

    def to_indices(self, tokens):
        single = isinstance(tokens, str)
        toks = [tokens] if single else tokens
        idx = [self._token_to_idx.get(t, 0) for t in toks]
        return idx[0] if single else idx

    def to_tokens(self, indices):
        single = isinstance(indices, int)
        idx = [indices] if single else indices
        for i in idx:
            if not 0 <= i < len(self._idx_to_token):
                raise ValueError(f"token index {i} out of range")
        toks = [self._idx_to_token[i] for i in idx]
        return toks[0] if single else toks
