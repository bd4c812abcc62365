"""Text helpers (reference parity: contrib/text/utils.py)."""
from __future__ import annotations

import collections
import re


def count_tokens_from_str(source_str, token_delim=" ", seq_delim="\n",
                          to_lower=False, counter_to_update=None):
    """Tokenize a string and count tokens (reference API)."""
    source_str = re.sub(f"[{token_delim}{seq_delim}]+", " ", source_str)
    if to_lower:
        source_str = source_str.lower()
    tokens = source_str.split()
    counter = counter_to_update if counter_to_update is not None \
        else collections.Counter()
    counter.update(tokens)
    return counter
