"""CNN tok2vec: MultiHashEmbed + MaxoutWindowEncoder.

Behavioral contract of spaCy's default tok2vec: 4 attrs
(NORM/PREFIX/SUFFIX/SHAPE) each hash-embedded (4 murmur rows summed), the
concat mixed by a Maxout+LN down to `width`, then `depth` residual blocks of
seq2col(window=1) -> Maxout -> LayerNorm.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

from spacy_ray_amd.ops import api as ops
from .batch import TokenBatch
from .layers import LayerNorm, Maxout


class MultiHashEmbed(nn.Module):
    def __init__(
        self,
        width: int,
        rows: List[int] = (5000, 2500, 2500, 2500),
        attrs: List[str] = ("NORM", "PREFIX", "SUFFIX", "SHAPE"),
        seed: int = 0,
    ):
        super().__init__()
        assert len(rows) == len(attrs)
        self.width = width
        self.attrs = list(attrs)
        self.seeds = [seed + i for i in range(len(attrs))]
        self.tables = nn.ParameterList(
            nn.Parameter(torch.randn(r, width) * (1.0 / (width ** 0.5)))
            for r in rows
        )
        self.mixer = Maxout(width * len(attrs), width, pieces=3, normalize=True)

    def forward(self, batch: TokenBatch, drop: float = 0.0) -> torch.Tensor:
        from spacy_ray_amd.utils import timing

        with timing.phase("t2v/embed_tables"):
            X = ops.multi_hashembed(batch.attr_ids, self.seeds, list(self.tables))
        with timing.phase("t2v/mixer_gemm"):
            Y = ops.linear_cdw(X, self.mixer.weight, self.mixer.bias)
        with timing.phase("t2v/mixer_maxout"):
            Y = ops.maxout(Y.view(*Y.shape[:-1], self.mixer.pieces, self.mixer.nO))
        with timing.phase("t2v/mixer_ln"):
            Y = self.mixer.norm(Y)
        with timing.phase("t2v/mixer_drop"):
            if drop and self.training:
                Y = torch.nn.functional.dropout(Y, drop)
        return Y


class MaxoutWindowEncoder(nn.Module):
    def __init__(self, width: int, depth: int = 4, window_size: int = 1, maxout_pieces: int = 3):
        super().__init__()
        assert window_size == 1, "window_size=1 is the supported CNN window"
        self.width = width
        self.depth = depth
        self.blocks = nn.ModuleList(
            Maxout(width * 3, width, pieces=maxout_pieces, normalize=True)
            for _ in range(depth)
        )

    def forward(self, X: torch.Tensor, lengths: torch.Tensor, drop: float = 0.0) -> torch.Tensor:
        from spacy_ray_amd.utils import timing

        # doc-boundary masks computed once for all depth layers (and reused
        # by every seq2col forward AND backward — no per-call index work)
        starts, ends = ops.boundary_masks_u8(lengths, X.shape[0])
        fused = ops.mwe_layer_available(X, self.width, self.blocks[0].pieces)
        for bi, block in enumerate(self.blocks):
            with timing.phase(f"t2v/encode{bi}"):
                if fused:
                    # whole block in ONE MFMA kernel (incl. dropout+residual)
                    mask = None
                    if drop and self.training:
                        mask = ops.dropout_mask_like(X, drop)
                    X = ops.mwe_layer(
                        X, block.weight, block.bias,
                        block.norm.weight, block.norm.bias, starts, ends,
                        mask, block.norm.eps,
                    )
                    continue
                Y = block(ops.seq2col(X, lengths, starts, ends))
                if drop and self.training:
                    Y = torch.nn.functional.dropout(Y, drop)
                X = X + Y  # residual (thinc `residual(...)` wrapper)
        return X


class Tok2Vec(nn.Module):
    """embed -> encode; output width = encode.width."""

    def __init__(self, embed: MultiHashEmbed, encode: MaxoutWindowEncoder):
        super().__init__()
        self.embed = embed
        self.encode = encode
        self.width = encode.width

    def forward(self, batch: TokenBatch, drop: float = 0.0) -> torch.Tensor:
        X = self.embed(batch, drop=drop)
        return self.encode(X, batch.lengths, drop=drop)
