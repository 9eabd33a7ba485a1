"""StringStore: interned string <-> 64-bit hash mapping.

Behavioral contract of spaCy's Cython StringStore:
strings map to stable uint64 hashes; the store keeps the reverse mapping for
strings it has seen.  Hash = murmur3 x64_128 low word, seed 1 (C++ core)."""
from __future__ import annotations

from typing import Dict, Iterable, List

import numpy as np

from spacy_ray_amd import _srx_cpu


class StringStore:
    def __init__(self, strings: Iterable[str] = ()) -> None:
        self._map: Dict[int, str] = {}
        for s in strings:
            self.add(s)

    def add(self, string: str) -> int:
        h = int(_srx_cpu.hash_string(string))
        self._map[h] = string
        return h

    def add_batch(self, strings: List[str]) -> np.ndarray:
        hashes = _srx_cpu.hash_strings(strings)
        for h, s in zip(hashes.tolist(), strings):
            self._map[h] = s
        return hashes

    def __getitem__(self, key):
        if isinstance(key, str):
            return int(_srx_cpu.hash_string(key))
        return self._map[int(key)]

    def __contains__(self, key) -> bool:
        if isinstance(key, str):
            return int(_srx_cpu.hash_string(key)) in self._map
        return int(key) in self._map

    def __len__(self) -> int:
        return len(self._map)

    def to_list(self) -> List[str]:
        return sorted(self._map.values())
