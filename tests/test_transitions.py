"""Transition systems: oracle consistency and termination.

The key property (used for training): following any sequence of zero-cost
actions from the initial state reconstructs the gold parse / gold BILUO
exactly (dynamic-oracle soundness)."""
import numpy as np
import pytest

from spacy_ray_amd import _srx_cpu
from spacy_ray_amd.data.corpus import _random_projective_heads
import random


def _follow_oracle_parser(lengths, heads, labels, n_labels, prefer="first"):
    b = _srx_cpu.ArcEagerBatch(np.asarray(lengths, dtype=np.int32), n_labels)
    b.set_gold(np.asarray(heads, dtype=np.int32), np.asarray(labels, dtype=np.int32))
    for _ in range(4 * sum(lengths) + 16):
        final = b.is_final()
        if final.all():
            break
        costs = b.costs()
        valid = b.valid().astype(bool)
        cmin = costs.min(axis=1, keepdims=True)
        is_gold = (costs <= cmin + 1e-6) & valid
        if prefer == "first":
            actions = is_gold.argmax(axis=1).astype(np.int32)
        else:  # random zero-cost action — oracle must still be sound
            actions = np.array(
                [np.random.RandomState(_ + i).choice(np.nonzero(r)[0]) if r.any() else 0
                 for i, r in enumerate(is_gold)],
                dtype=np.int32,
            )
        actions[final.astype(bool)] = -1
        b.advance(actions)
    assert b.is_final().all()
    return b.heads(), b.labels()


def test_arc_eager_oracle_reconstructs_gold():
    rng = random.Random(42)
    lengths, heads, labels = [], [], []
    for n in (2, 3, 5, 8, 13, 20):
        h = _random_projective_heads(n, rng)
        lengths.append(n)
        heads.extend(h)
        labels.extend([(i * 7) % 5 for i in range(n)])
    ph, pl = _follow_oracle_parser(lengths, heads, labels, 5)
    gh = np.asarray(heads)
    assert (ph == gh).all(), (ph.tolist(), gh.tolist())
    # labels match wherever an arc exists (roots keep -1)
    gl = np.asarray(labels)
    mask = gh >= 0
    assert (pl[mask] == gl[mask]).all()


def test_arc_eager_oracle_random_tiebreak():
    rng = random.Random(7)
    for trial in range(5):
        n = rng.randint(3, 12)
        h = _random_projective_heads(n, rng)
        labels = [rng.randrange(3) for _ in range(n)]
        ph, _ = _follow_oracle_parser([n], h, labels, 3, prefer="random")
        assert (ph == np.asarray(h)).all()


def test_arc_eager_terminates_on_any_policy():
    # worst-case policy: always the first valid action
    n = 10
    b = _srx_cpu.ArcEagerBatch(np.array([n], dtype=np.int32), 2)
    for _ in range(4 * n + 16):
        if b.is_final().all():
            break
        valid = b.valid().astype(bool)
        assert valid.any(axis=1).all(), "non-final state with no valid action"
        actions = valid.argmax(axis=1).astype(np.int32)
        b.advance(actions)
    assert b.is_final().all()


def test_ner_oracle_reconstructs_gold():
    # gold: O B-1 I-1 L-1 U-0 O  (type ids: code = 1+4t+kind)
    codes = np.array([0, 1 + 4, 2 + 4, 3 + 4, 4, 0], dtype=np.int32)
    b = _srx_cpu.BiluoBatch(np.array([6], dtype=np.int32), 2)
    b.set_gold(codes)
    for _ in range(10):
        if b.is_final().all():
            break
        costs = b.costs()
        valid = b.valid().astype(bool)
        cmin = costs.min(axis=1, keepdims=True)
        is_gold = (costs <= cmin + 1e-6) & valid
        b.advance(is_gold.argmax(axis=1).astype(np.int32))
    assert b.is_final().all()
    assert (b.tags() == codes).all()


def test_ner_begin_invalid_at_last_token():
    b = _srx_cpu.BiluoBatch(np.array([1], dtype=np.int32), 1)
    v = b.valid()[0]
    # actions: [OUT, B, I, L, U] — only OUT and UNIT valid on a 1-token doc
    assert v.tolist() == [1, 0, 0, 0, 1]


def test_ner_open_entity_constraints():
    b = _srx_cpu.BiluoBatch(np.array([3], dtype=np.int32), 2)
    b.advance(np.array([1], dtype=np.int32))  # BEGIN type 0
    v = b.valid()[0]
    # open type 0: only IN(0)=2 and LAST(0)=3 valid
    assert v[2] == 1 and v[3] == 1
    assert v[0] == 0 and v[1] == 0 and v[4] == 0
    assert v[6] == 0 and v[7] == 0  # type-1 IN/LAST invalid
