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


def test_oracle_costs_match_bruteforce_reference():
    """Property test: the incremental O(1) oracle
    (on-stack bitmap + gold-children-in-buffer counters) must equal a
    brute-force Goldberg-Nivre cost computation at every state of random
    valid-action walks over random projective-ish gold trees."""
    import numpy as np
    from spacy_ray_amd import _srx_cpu

    rng = np.random.RandomState(7)
    for trial in range(40):
        n = int(rng.randint(2, 12))
        L = int(rng.randint(1, 4))
        heads = np.full(n, -1, dtype=np.int32)
        for i in range(1, n):
            heads[i] = rng.randint(0, i)  # head earlier than child (projective-ish)
        if n > 1 and rng.rand() < 0.5:
            heads[0] = rng.randint(1, n)
        labels = rng.randint(0, L, size=n).astype(np.int32)
        batch = _srx_cpu.ArcEagerBatch(np.array([n], dtype=np.int32), L, 0)
        batch.set_gold(heads, labels)
        A = batch.n_actions

        # python mirror of the state for brute-force costs
        stack, buf, head = [], 0, [-1] * n
        KINV = 1e9

        def brute_costs(valid):
            gh, gl = heads, labels
            b = buf if buf < n else -1
            s0 = stack[-1] if stack else -1
            c_shift = c_reduce = c_la = c_ra = 0.0
            if b >= 0:
                if gh[b] >= 0 and gh[b] in stack:
                    c_shift += 1
                c_shift += sum(1 for s in stack if head[s] == -1 and gh[s] == b)
            if s0 >= 0:
                c_reduce = sum(1 for t in range(buf, n) if gh[t] == s0)
                if b >= 0:
                    c_la = c_reduce
                    if gh[s0] >= 0 and gh[s0] > b:
                        c_la += 1
                    if gh[b] >= 0 and gh[b] != s0 and (gh[b] in stack or gh[b] > b):
                        c_ra += 1
                    c_ra += sum(1 for s in stack if head[s] == -1 and gh[s] == b)
            out = np.full(A, KINV, dtype=np.float32)
            if valid[0]:
                out[0] = c_shift
            if valid[1]:
                out[1] = c_reduce
            for l in range(L):
                la, ra = c_la, c_ra
                if b >= 0 and s0 >= 0:
                    if gh[s0] == b and gl[s0] != l:
                        la += 1
                    if gh[b] == s0 and gl[b] != l:
                        ra += 1
                if valid[2 + l]:
                    out[2 + l] = la
                if valid[2 + L + l]:
                    out[2 + L + l] = ra
            return out

        for _ in range(4 * n):
            fin = batch.is_final()
            if fin[0]:
                break
            valid = batch.valid()[0]
            costs = batch.costs()[0]
            ref = brute_costs(valid)
            assert np.allclose(costs, ref), (trial, costs, ref, stack, buf, head)
            choices = np.nonzero(valid)[0]
            act = int(choices[rng.randint(len(choices))])
            batch.advance(np.array([act], dtype=np.int32))
            # python mirror advance
            if act == 0:
                stack.append(buf); buf += 1
            elif act == 1:
                stack.pop()
            elif act < 2 + L:
                head[stack[-1]] = buf; stack.pop()
            else:
                head[buf] = stack[-1]; stack.append(buf); buf += 1


def test_break_action_sentence_boundaries():
    """BREAK (use_break=True): valid only with an empty stack and consumed
    prefix; oracle prefers it exactly at gold sentence starts; predicted
    boundaries come back via sent_starts()."""
    import numpy as np
    from spacy_ray_amd import _srx_cpu

    # two sentences of 2 tokens each: heads [1,-1, 3,-1] (heads within sent)
    n, L = 4, 1
    batch = _srx_cpu.ArcEagerBatch(np.array([n], dtype=np.int32), L, 0, True)
    A = batch.n_actions
    assert A == 2 + 2 * L + 1
    BREAK = A - 1
    heads = np.array([1, -1, 3, -1], dtype=np.int32)
    labels = np.zeros(n, dtype=np.int32)
    batch.set_gold(heads, labels)
    batch.set_sent_gold(np.array([1, 0, 1, 0], dtype=np.int32))
    # initial state: stack empty, buf=0 -> BREAK invalid (nothing consumed)
    assert batch.valid()[0][BREAK] == 0
    # follow min-cost actions greedily; collect them
    taken = []
    for _ in range(4 * n):
        if batch.is_final()[0]:
            break
        costs = batch.costs()[0]
        act = int(np.argmin(costs))
        taken.append(act)
        batch.advance(np.array([act], dtype=np.int32))
    assert BREAK in taken  # the oracle used the boundary
    sents = batch.sent_starts()
    assert sents[2] == 1  # token 2 marked as a sentence start
    assert sents[1] == 0 and sents[3] == 0
    # parse quality unaffected by the break
    assert batch.heads()[0] == 1 and batch.heads()[2] == 3


def test_break_never_preferred_without_sentence_gold():
    import numpy as np
    from spacy_ray_amd import _srx_cpu

    batch = _srx_cpu.ArcEagerBatch(np.array([4], dtype=np.int32), 1, 0, True)
    BREAK = batch.n_actions - 1
    batch.set_gold(np.array([1, -1, 1, 1], dtype=np.int32),
                   np.zeros(4, dtype=np.int32))
    for _ in range(16):
        if batch.is_final()[0]:
            break
        costs = batch.costs()[0]
        act = int(np.argmin(costs))
        assert act != BREAK  # cost 1 > some zero-cost action
        batch.advance(np.array([act], dtype=np.int32))
    assert batch.sent_starts().sum() == 0


def test_pack_step_gold_mask_matches_bruteforce_costs():
    """pack_step's scalar-cost min-cost mask must equal the mask derived
    from the full costs() row at every state of random walks (with and
    without BREAK)."""
    import numpy as np
    from spacy_ray_amd import _srx_cpu

    rng = np.random.RandomState(11)
    for trial in range(30):
        n = int(rng.randint(2, 12))
        L = int(rng.randint(1, 5))
        use_break = bool(trial % 2)
        heads = np.full(n, -1, dtype=np.int32)
        for i in range(1, n):
            heads[i] = rng.randint(0, i)
        labels = rng.randint(0, L, size=n).astype(np.int32)
        batch = _srx_cpu.ArcEagerBatch(np.array([n], dtype=np.int32), L, 0,
                                       use_break)
        batch.set_gold(heads, labels)
        if use_break:
            sents = (rng.rand(n) < 0.3).astype(np.int32)
            sents[0] = 1
            batch.set_sent_gold(sents)
        for _ in range(4 * n):
            if batch.is_final()[0]:
                break
            act_idx, feats, valid, gold = batch.step_arrays(True)
            costs = batch.costs()[0]
            vmask = batch.valid()[0]
            cmin = costs[vmask > 0].min()
            ref_gold = ((vmask > 0) & (costs <= cmin + 1e-6)).astype(np.uint8)
            assert np.array_equal(valid[0], vmask), (trial, valid[0], vmask)
            assert np.array_equal(gold[0], ref_gold), (
                trial, gold[0], ref_gold, costs)
            choices = np.nonzero(ref_gold)[0]
            act = int(choices[rng.randint(len(choices))])
            batch.advance(np.array([act], dtype=np.int32))


def test_beam_python_states_mirror_cpp_systems():
    """The beam decoder's pure-python states must track the C++ transition
    systems exactly (features, valid masks, finality, final heads/labels /
    tags) through random valid action sequences."""
    import random

    from spacy_ray_amd.pipeline.beam import _ArcEagerState, _BiluoState

    rng = random.Random(0)
    L = 3  # labels / types
    for trial in range(40):
        n = rng.randint(1, 14)
        cpp = _srx_cpu.ArcEagerBatch(np.array([n], dtype=np.int32), L)
        py = _ArcEagerState(n)
        for _ in range(2 * n + 2):
            assert bool(cpp.is_final()[0]) == py.is_final()
            if py.is_final():
                break
            v_cpp = cpp.valid()[0].astype(bool)
            v_py = py.valid_mask(L)
            assert (v_cpp == v_py).all(), (trial, v_cpp, v_py)
            f_cpp = cpp.features()[0]
            f_py = np.asarray(py.features(), dtype=np.int32)
            assert (f_cpp == f_py).all(), (trial, f_cpp, f_py)
            act = rng.choice(np.flatnonzero(v_py))
            cpp.advance(np.array([act], dtype=np.int32))
            py.apply(int(act), L)
        assert (cpp.heads() == np.asarray(py.head, dtype=np.int32)).all()
        assert (cpp.labels() == np.asarray(py.label, dtype=np.int32)).all()
    for trial in range(40):
        n = rng.randint(1, 14)
        cpp = _srx_cpu.BiluoBatch(np.array([n], dtype=np.int32), L)
        py = _BiluoState(n)
        for _ in range(n + 1):
            assert bool(cpp.is_final()[0]) == py.is_final()
            if py.is_final():
                break
            v_cpp = cpp.valid()[0].astype(bool)
            v_py = py.valid_mask(L)
            assert (v_cpp == v_py).all(), (trial, v_cpp, v_py)
            f_cpp = cpp.features()[0]
            f_py = np.asarray(py.features(), dtype=np.int32)
            assert (f_cpp == f_py).all(), (trial, f_cpp, f_py)
            act = rng.choice(np.flatnonzero(v_py))
            cpp.advance(np.array([act], dtype=np.int32))
            py.apply(int(act), L)
