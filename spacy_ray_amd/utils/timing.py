"""Per-step phase timers (SURVEY.md §5.1 disposition: data / fwd / bwd /
comm / opt wall-clock, aggregated per phase).

Enabled with SRX_TIMING=1 (adds torch.cuda.synchronize at phase edges — for
diagnosis, not for production runs).  `phase_times()` returns cumulative ms
per label; `reset_times()` clears."""
from __future__ import annotations

import os
import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict

import torch

_ENABLED = os.environ.get("SRX_TIMING") == "1"
_times: Dict[str, float] = defaultdict(float)
_counts: Dict[str, int] = defaultdict(int)


def enabled() -> bool:
    return _ENABLED


@contextmanager
def phase(name: str):
    if not _ENABLED:
        yield
        return
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    try:
        yield
    finally:
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        _times[name] += (time.perf_counter() - t0) * 1000
        _counts[name] += 1


@contextmanager
def span(name: str):
    """Wall-clock span WITHOUT cuda sync (for inner-loop spans where a sync
    would perturb the measurement; GPU waits show up in whichever span
    contains the synchronizing call)."""
    if not _ENABLED:
        yield
        return
    t0 = time.perf_counter()
    try:
        yield
    finally:
        _times[name] += (time.perf_counter() - t0) * 1000
        _counts[name] += 1


def phase_times() -> Dict[str, float]:
    return dict(_times)


def phase_counts() -> Dict[str, int]:
    return dict(_counts)


def reset_times() -> None:
    _times.clear()
    _counts.clear()
