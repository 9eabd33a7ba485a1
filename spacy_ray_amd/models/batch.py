"""TokenBatch: the device-side representation of a batch of Docs.

MI355X-first design (SURVEY.md §2.2 N6 disposition): attr hashes are
precomputed per Doc on the CPU once, and a batch ships to the GPU as ONE
int64 [T, 4] tensor (bit-cast uint64) + a lengths vector — no per-token
Python objects anywhere near the hot path.
"""
from __future__ import annotations

from typing import Sequence

import numpy as np
import torch

from spacy_ray_amd.vocab.doc import Doc


class TokenBatch:
    __slots__ = ("attr_ids", "lengths", "n_tokens", "n_real_tokens", "docs",
                 "staged", "lengths_np")

    def __init__(self, docs: Sequence[Doc], device: torch.device,
                 pad_to: int = 0):
        """pad_to > 0: round the token count up to a multiple by appending a
        pad pseudo-doc (zero attr ids, no annotations).  Keeps the GEMM M
        dims on a small set of repeating shapes so hipBLASLt's solution
        cache hits instead of re-selecting per batch.  Defaults: 2048 on
        GPU, none on CPU."""
        if isinstance(device, str):
            device = torch.device(device)
        self.docs = list(docs)
        lens_list = [len(d) for d in docs]
        # real (pre-pad) per-doc lengths; reused by the transition pipes'
        # state construction (a fresh 50k-doc list comprehension per pipe
        # per step showed up in profiles)
        self.lengths_np = np.asarray(lens_list, dtype=np.int32)
        real = int(sum(lens_list))
        self.n_real_tokens = real
        if pad_to == 0:
            if device.type == "cuda":
                # coarser buckets at larger T: hipBLASLt re-runs solution
                # selection per distinct M, so keep the distinct-M count tiny
                pad_to = 2048 if real < 32768 else 8192
            else:
                pad_to = 1
        padded = -(-max(real, 1) // pad_to) * pad_to if pad_to > 1 else real
        n_pad = padded - real
        if n_pad > 0:
            lens_list = lens_list + [n_pad]
        lens = np.array(lens_list, dtype=np.int64)
        arrs = [d.attr_hashes for d in docs]
        if n_pad > 0:
            arrs.append(np.zeros((n_pad, 4), dtype=np.uint64))
        attr = (np.concatenate(arrs, axis=0) if arrs
                else np.zeros((0, 4), dtype=np.uint64))
        self.n_tokens = int(lens.sum())
        from spacy_ray_amd.utils.pinned import to_device

        # bit-cast uint64 -> int64 (torch has no uint64); kernels re-interpret
        self.attr_ids = to_device(attr.view(np.int64), device)
        self.lengths = to_device(lens, device)
        # per-pipe staged device data (e.g. tagger gold ids), uploaded at
        # step start while the GPU queue is empty — a pageable H2D later in
        # the step blocks the host behind every queued kernel.  Replayed
        # batches (bench) keep the cache across steps.
        self.staged = {}

    def __len__(self) -> int:
        return len(self.docs)
