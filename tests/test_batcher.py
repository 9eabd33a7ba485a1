"""Batcher semantics (spaCy contracts: word budget + tolerance, oversize
handling, compounding schedule sizes)."""
from spacy_ray_amd.config.registry import registry

registry.ensure_populated()


def _words(n):
    return ["w"] * n


def test_batch_by_words_respects_budget_and_tolerance():
    b = registry.batchers.get("spacy.batch_by_words.v1")(size=10, tolerance=0.2)
    items = [_words(n) for n in (4, 4, 4, 4, 4)]
    batches = list(b(items))
    assert [sum(len(x) for x in bt) for bt in batches] == [12, 8]
    assert all(sum(len(x) for x in bt) <= 12 for bt in batches)


def test_batch_by_words_oversize_yields_alone_or_discards():
    b = registry.batchers.get("spacy.batch_by_words.v1")(size=10, tolerance=0.0)
    items = [_words(3), _words(25), _words(3)]
    batches = list(b(items))
    assert [len(x) for bt in batches for x in bt].count(25) == 1  # yielded alone
    b2 = registry.batchers.get("spacy.batch_by_words.v1")(
        size=10, tolerance=0.0, discard_oversize=True)
    batches2 = list(b2(items))
    assert all(len(x) <= 10 for bt in batches2 for x in bt)  # dropped


def test_batch_by_words_compounding_schedule():
    from spacy_ray_amd.config.config import resolve

    schedule = resolve({"@schedules": "compounding.v1", "start": 4, "stop": 16,
                        "compound": 2.0})
    b = registry.batchers.get("spacy.batch_by_words.v1")(size=schedule,
                                                         tolerance=0.0)
    items = [_words(4)] * 8
    batches = list(b(items))
    counts = [sum(len(x) for x in bt) for bt in batches]
    assert counts[0] == 4           # first batch at start size
    assert counts[-1] >= counts[0]  # budget grows


def test_batch_by_sequence_counts():
    b = registry.batchers.get("spacy.batch_by_sequence.v1")(size=3)
    batches = list(b(list(range(8))))
    assert [len(bt) for bt in batches] == [3, 3, 2]


def test_batch_by_padded_size_bound():
    b = registry.batchers.get("spacy.batch_by_padded.v1")(size=20)
    items = [_words(n) for n in (5, 5, 5, 9, 2)]
    for bt in b(items):
        assert max(len(x) for x in bt) * len(bt) <= 20


def test_schedules_contracts():
    from spacy_ray_amd.config.config import resolve

    wl = resolve({"@schedules": "warmup_linear.v1", "initial_rate": 0.1,
                  "warmup_steps": 10, "total_steps": 110})
    assert wl(0) == 0.01                      # linear ramp
    assert wl(9) == 0.1                       # peak at end of warmup
    assert abs(wl(60) - 0.05) < 1e-9          # halfway down
    assert wl(110) == 0.0 and wl(1000) == 0.0

    comp = resolve({"@schedules": "compounding.v1", "start": 2.0, "stop": 16.0,
                    "compound": 2.0})
    assert [comp(i) for i in range(5)] == [2.0, 4.0, 8.0, 16.0, 16.0]
    # iterator protocol (batchers call next())
    it = iter(comp)
    assert [next(it) for _ in range(4)] == [2.0, 4.0, 8.0, 16.0]

    const = resolve({"@schedules": "constant.v1", "rate": 0.3})
    assert const(0) == const(999) == 0.3
