"""Optimizer config (@optimizers registry) + single-process Adam wrapper.

Thinc ``Adam.v1`` contract (resolved from [training.optimizer],
`/root/reference/spacy_ray/worker.py:245` hands it to the proxy which runs it
per owned shard, `/root/reference/spacy_ray/proxies.py:128`).  Here the spec
is consumed either by SimpleAdam (1-process) or by the flat sharded-Adam
engine in parallel/ (ZeRO-1 over RCCL).

Divergences from thinc, by design: grad clipping is global-norm over the
whole flat gradient (thinc clips per tensor) — one fused kernel over the
flat buffer; `use_averages` keeps a running parameter mean that is swapped in for eval and checkpoints (engine.averaged_params).
"""
from __future__ import annotations

from typing import Optional, Union

import torch

from spacy_ray_amd.config.registry import registry
from .schedules import Schedule


class AdamSpec:
    def __init__(self, learn_rate, beta1, beta2, eps, L2, grad_clip,
                 L2_is_weight_decay, use_averages):
        self.learn_rate = learn_rate
        self.beta1 = beta1
        self.beta2 = beta2
        self.eps = eps
        self.L2 = L2
        self.grad_clip = grad_clip
        self.L2_is_weight_decay = L2_is_weight_decay
        self.use_averages = use_averages

    def lr(self, step: int) -> float:
        if isinstance(self.learn_rate, Schedule):
            return float(self.learn_rate(step))
        if callable(self.learn_rate):
            return float(self.learn_rate(step))
        return float(self.learn_rate)


@registry.optimizers("Adam.v1")
def make_adam(
    learn_rate: Union[float, Schedule] = 0.001,
    beta1: float = 0.9,
    beta2: float = 0.999,
    eps: float = 1e-8,
    L2: float = 0.01,
    grad_clip: float = 1.0,
    L2_is_weight_decay: bool = True,
    use_averages: bool = False,
) -> AdamSpec:
    return AdamSpec(learn_rate, beta1, beta2, eps, L2, grad_clip,
                    L2_is_weight_decay, use_averages)


class SimpleAdam:
    """Single-process optimizer over a torch module (CPU tests / smoke)."""

    def __init__(self, module: torch.nn.Module, spec: AdamSpec):
        self.spec = spec
        self.params = [p for p in module.parameters() if p.requires_grad]
        self.opt = torch.optim.AdamW(
            self.params,
            lr=spec.lr(0),
            betas=(spec.beta1, spec.beta2),
            eps=spec.eps,
            weight_decay=spec.L2 if spec.L2_is_weight_decay else 0.0,
        )
        self.step_count = 0

    def step(self) -> None:
        if self.spec.grad_clip:
            torch.nn.utils.clip_grad_norm_(self.params, self.spec.grad_clip)
        for g in self.opt.param_groups:
            g["lr"] = self.spec.lr(self.step_count)
        self.opt.step()
        self.step_count += 1

    def zero_grad(self) -> None:
        self.opt.zero_grad(set_to_none=False)
