"""TokenBatch: the device-side representation of a batch of Docs.

MI355X-first design (SURVEY.md §2.2 N6 disposition): attr hashes are
precomputed per Doc on the CPU once, and a batch ships to the GPU as ONE
int64 [T, 4] tensor (bit-cast uint64) + a lengths vector — no per-token
Python objects anywhere near the hot path.
"""
from __future__ import annotations

from typing import List, Sequence

import numpy as np
import torch

from spacy_ray_amd.vocab.doc import Doc


class TokenBatch:
    __slots__ = ("attr_ids", "lengths", "n_tokens", "docs")

    def __init__(self, docs: Sequence[Doc], device: torch.device):
        self.docs = list(docs)
        lens = np.array([len(d) for d in docs], dtype=np.int64)
        if len(docs):
            attr = np.concatenate([d.attr_hashes for d in docs], axis=0)
        else:
            attr = np.zeros((0, 4), dtype=np.uint64)
        self.n_tokens = int(lens.sum())
        # bit-cast uint64 -> int64 (torch has no uint64); kernels re-interpret
        self.attr_ids = torch.from_numpy(attr.view(np.int64)).to(device, non_blocking=True)
        self.lengths = torch.from_numpy(lens).to(device, non_blocking=True)

    def __len__(self) -> int:
        return len(self.docs)
