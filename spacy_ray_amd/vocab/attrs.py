"""Token attribute extraction: NORM / PREFIX / SUFFIX / SHAPE.

These are the four attrs the default MultiHashEmbed embeds (behavioral
contract of spaCy's lex_attrs + MultiHashEmbed.v2).
Each attr value is a string mapped to a uint64 murmur hash; the embedding
kernels re-hash those uint64 ids into table rows.
"""
from __future__ import annotations

from typing import List

import numpy as np

from spacy_ray_amd import _srx_cpu

ATTR_NAMES = ("NORM", "PREFIX", "SUFFIX", "SHAPE")
N_ATTRS = len(ATTR_NAMES)


def word_shape(text: str) -> str:
    """spaCy-style shape: letters -> X/x, digits -> d, else kept; runs longer
    than 4 are truncated to 4 (contract of spacy.lang.lex_attrs.word_shape)."""
    if len(text) >= 100:
        return "LONG"
    shape: List[str] = []
    last = ""
    run = 0
    for ch in text:
        if ch.isalpha():
            s = "X" if ch.isupper() else "x"
        elif ch.isdigit():
            s = "d"
        else:
            s = ch
        if s == last:
            run += 1
        else:
            run = 1
            last = s
        if run < 5:
            shape.append(s)
    return "".join(shape)


def attr_strings(word: str) -> List[str]:
    return [word.lower(), word[:1], word[-3:], word_shape(word)]


def extract_attr_hashes(words: List[str]) -> np.ndarray:
    """(n_tokens, 4) uint64 — one murmur-hashed value per attr per token."""
    n = len(words)
    flat: List[str] = []
    for w in words:
        flat.extend(attr_strings(w))
    hashes = _srx_cpu.hash_strings(flat)
    return hashes.reshape(n, N_ATTRS) if n else np.zeros((0, N_ATTRS), dtype=np.uint64)
