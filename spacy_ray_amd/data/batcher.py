"""Batching: batch_by_words with tolerance + oversize handling.

Contract of spaCy's batcher resolved from ``[training.batcher]``
(`/root/reference/spacy_ray/worker.py:173` passes T["batcher"] into
create_train_batches).  ``size`` may be an int or a schedule (generator /
callable from @schedules).
"""
from __future__ import annotations

from typing import Any, Callable, Iterable, Iterator, List, Sequence

from spacy_ray_amd.config.registry import registry


def _size_iter(size: Any) -> Iterator[float]:
    if hasattr(size, "__next__"):
        return size
    if callable(size):  # schedule function: step -> value
        def gen():
            step = 0
            while True:
                yield size(step)
                step += 1
        return gen()
    def const():
        while True:
            yield size
    return const()


@registry.batchers("spacy.batch_by_words.v1")
def configure_minibatch_by_words(
    size: Any = 5000,
    tolerance: float = 0.2,
    discard_oversize: bool = False,
    get_length: Callable = len,
):
    sizes = _size_iter(size)

    def batcher(items: Iterable) -> Iterator[List]:
        target = next(sizes)
        max_size = target + target * tolerance
        batch: List = []
        n_words = 0.0
        for item in items:
            n = get_length(item)
            if n == 0:
                continue
            if n > max_size:
                if not discard_oversize:
                    yield [item]
                    target = next(sizes)
                    max_size = target + target * tolerance
                continue
            if n_words + n > max_size and batch:
                yield batch
                target = next(sizes)
                max_size = target + target * tolerance
                batch = []
                n_words = 0.0
            batch.append(item)
            n_words += n
        if batch:
            yield batch

    return batcher


@registry.batchers("spacy.batch_by_sequence.v1")
def configure_minibatch(size: Any = 64, get_length: Callable = len):
    sizes = _size_iter(size)

    def batcher(items: Iterable) -> Iterator[List]:
        n = int(next(sizes))
        batch: List = []
        for item in items:
            batch.append(item)
            if len(batch) >= n:
                yield batch
                n = int(next(sizes))
                batch = []
        if batch:
            yield batch

    return batcher


@registry.batchers("spacy.batch_by_padded.v1")
def configure_minibatch_by_padded_size(
    size: Any = 8000,
    buffer: int = 256,
    discard_oversize: bool = False,
    get_length: Callable = len,
):
    """Batch so that padded size (max_len * n_seqs) stays under `size`."""
    sizes = _size_iter(size)

    def batcher(items: Iterable) -> Iterator[List]:
        target = next(sizes)
        batch: List = []
        max_len = 0
        for item in items:
            n = get_length(item)
            new_max = max(max_len, n)
            if new_max * (len(batch) + 1) > target and batch:
                yield batch
                target = next(sizes)
                batch = []
                max_len = 0
                new_max = n
            if n > target:
                if not discard_oversize:
                    yield [item]
                    target = next(sizes)
                continue
            batch.append(item)
            max_len = new_max
        if batch:
            yield batch

    return batcher
