"""Training loop with the ``train_while_improving`` contract.

Mirrors the step-iterator semantics the reference consumes at
`/root/reference/spacy_ray/worker.py:176-189` (spaCy training/loop.py [dep]):
yields (batch, info, is_best_checkpoint) per step; evaluates every
``eval_frequency`` steps; stops on patience / max_steps / data exhaustion.
The reference hard-codes accumulate_gradient=1 into the loop while reading
the config key — here it is wired properly: each batch is
split into ``accumulate_gradient`` sub-batches whose gradients accumulate
before one optimizer step.
"""
from __future__ import annotations

import random
from typing import Callable, Dict, Iterable, Iterator, List, Optional, Sequence

from spacy_ray_amd.vocab.doc import Example


def create_train_batches(nlp, corpus, batcher, max_epochs: int):
    """Epoch-looped batch iterator (contract of spaCy create_train_batches,
    used at `/root/reference/spacy_ray/worker.py:170-175`).

    Deliberate difference from spaCy: per-epoch shuffling lives in the
    corpus reader (`SyntheticCorpus(shuffle=...)` / user readers), not
    here — the corpus is re-streamed each epoch (spaCy's max_epochs=-1
    behavior) instead of materialized + random.shuffle'd, which keeps
    epoch order independent of global RNG state across ranks."""
    epoch = 0
    while max_epochs < 1 or epoch < max_epochs:
        examples = corpus(nlp)
        count = 0
        for batch in batcher(examples):
            count += 1
            yield epoch, batch
        if count == 0:
            raise ValueError("empty training corpus")
        epoch += 1


def _subdivide(batch: List[Example], n: int) -> List[List[Example]]:
    if n <= 1 or len(batch) <= 1:
        return [batch]
    size = (len(batch) + n - 1) // n
    return [batch[i : i + size] for i in range(0, len(batch), size)]


def train_while_improving(
    nlp,
    stepper,
    train_data: Iterator,
    *,
    evaluate: Callable[[], tuple],
    dropout: float = 0.1,
    accumulate_gradient: int = 1,
    patience: int = 0,
    max_steps: int = 0,
    eval_frequency: int = 200,
    exclude: Sequence[str] = (),
    annotating_components: Sequence[str] = (),
    before_update: Optional[Callable] = None,
    initial_best: Optional[float] = None,
):
    """Generator of (batch, info, is_best_checkpoint).

    `stepper` abstracts the optimizer/comm engine:
        stepper.accumulate(examples, drop, losses) -> None  (fwd+bwd)
        stepper.apply_step() -> None                        (clip+opt+sync)
    `evaluate()` -> (score, other_scores).
    """
    # resume: seed the best-so-far so a resumed run cannot overwrite
    # model-best with a worse model (step -1 = "before this run")
    results = [] if initial_best is None else [(float(initial_best), -1)]
    losses: Dict[str, float] = {}
    words_seen = 0
    nlp._frozen = list(exclude)
    nlp._annotating = list(annotating_components)
    for step, (epoch, batch) in enumerate(train_data):
        if before_update is not None:
            before_update(nlp, {"step": step, "epoch": epoch})
        n_words = sum(len(eg) for eg in batch)
        words_seen += n_words
        subs = _subdivide(batch, accumulate_gradient)
        for i, sub in enumerate(subs):
            stepper.accumulate(sub, drop=dropout, losses=losses, sync=(i == len(subs) - 1))
        stepper.apply_step()
        if (step % eval_frequency) == 0 and step > 0 or (
            eval_frequency == 1 and step == 0
        ):
            score, other_scores = evaluate()
            results.append((score, step))
            best_score = max(r[0] for r in results)
            is_best = score >= best_score
        else:
            score, other_scores, is_best = None, {}, None
        info = {
            "epoch": epoch,
            "step": step,
            "score": score,
            "other_scores": other_scores,
            "losses": dict(losses),
            "checkpoints": list(results),
            "words": n_words,
            "words_seen": words_seen,
        }
        yield batch, info, is_best
        if is_best is not None:
            losses = {}
        if max_steps and step >= max_steps - 1:
            break
        if patience and results:
            # tuple max: ties prefer the LATER step (spaCy semantics),
            # so a plateau doesn't count against patience from its start.
            # patience is in STEPS, exactly as in spaCy's train_while_improving
            # (a patience=1600 / eval_frequency=200 config stops after 1600
            # stagnant steps, i.e. 8 evals).
            best_step = max(results)[1]
            if (step - best_step) >= patience:
                break
    nlp._frozen = []
    nlp._annotating = []
