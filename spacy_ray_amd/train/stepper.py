"""Steppers: the seam between the training loop and the optimizer/comm engine.

The loop calls accumulate() (forward+backward) then apply_step() (clip +
optimizer + any collective sync).  SimpleStepper is the 1-process CPU/GPU
path; the distributed ZeRO-1 engine in parallel/engine.py implements the same
interface (the injectable-backend seam idea the reference gets from its
``ray=`` constructor args)."""
from __future__ import annotations

from typing import Dict, Optional

import torch

from spacy_ray_amd.train.optimizer import AdamSpec, SimpleAdam


class SimpleStepper:
    def __init__(self, nlp, spec: AdamSpec):
        self.nlp = nlp
        self.module = nlp.torch_module()
        self.opt = SimpleAdam(self.module, spec)

    def accumulate(self, examples, drop: float = 0.0, losses: Optional[Dict] = None,
                   sync: bool = True) -> None:
        total, _ = self.nlp.forward_loss(examples, losses=losses, drop=drop)
        total.backward()
        if losses is not None:  # deferred display-loss tensors -> floats
            for k, v in losses.items():
                if torch.is_tensor(v):
                    losses[k] = float(v)

    def apply_step(self) -> None:
        self.opt.step()
        self.opt.zero_grad()
