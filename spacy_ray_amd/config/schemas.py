"""Typed schemas for config resolution (pydantic).

The reference resolves ``[training]`` against spaCy's ConfigSchemaTraining
(`/root/reference/spacy_ray/worker.py:93`); this is the equivalent typed
surface for the keys this engine consumes (the SURVEY.md §5.6 key list).
Unknown keys are allowed through (spaCy configs carry extras like
``seed``/``gpu_allocator`` consumers resolve separately)."""
from __future__ import annotations

from typing import Any, Dict, List, Optional

from pydantic import BaseModel, ConfigDict, field_validator


class ConfigSchemaTraining(BaseModel):
    model_config = ConfigDict(extra="allow", arbitrary_types_allowed=True)

    train_corpus: str = "corpora.train"
    dev_corpus: str = "corpora.dev"
    seed: Optional[int] = 0
    dropout: float = 0.1
    accumulate_gradient: int = 1
    patience: int = 0
    max_epochs: int = 0
    max_steps: int = 0
    eval_frequency: int = 200
    frozen_components: List[str] = []
    annotating_components: List[str] = []
    before_update: Optional[Any] = None
    before_to_disk: Optional[Any] = None
    gpu_allocator: Optional[str] = None
    logger: Optional[Any] = None
    batcher: Optional[Any] = None
    optimizer: Optional[Any] = None
    score_weights: Dict[str, Optional[float]] = {}

    @field_validator("accumulate_gradient")
    @classmethod
    def _pos_accum(cls, v: int) -> int:
        if v < 1:
            raise ValueError("accumulate_gradient must be >= 1")
        return v

    @field_validator("dropout")
    @classmethod
    def _dropout_range(cls, v: float) -> float:
        if not 0.0 <= v < 1.0:
            raise ValueError("dropout must be in [0, 1)")
        return v
