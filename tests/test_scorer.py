"""Direct scorer unit tests on hand-constructed examples (the convergence
tests exercise the scorer indirectly; these pin the exact math)."""
import pytest

from spacy_ray_amd.train.scorer import _ents_to_spans, score_examples, weighted_score
from spacy_ray_amd.vocab.doc import Doc, Example, Vocab


def _example(words, gold, pred):
    v = Vocab()
    ref = Doc(v, words, **gold)
    hyp = Doc(v, words, **pred)
    eg = Example.from_doc(ref)
    eg.predicted = hyp
    return eg


def test_tag_accuracy_math():
    eg = _example(["a", "b", "c", "d"],
                  dict(tags=["X", "Y", "X", "Y"]),
                  dict(tags=["X", "Y", "Y", "Y"]))
    s = score_examples([eg], ["tagger"])
    assert s["tag_acc"] == pytest.approx(3 / 4)


def test_uas_las_math():
    eg = _example(
        ["a", "b", "c"],
        dict(heads=[1, -1, 1], deps=["d1", "root", "d2"]),
        dict(heads=[1, -1, 0], deps=["dX", "root", "d2"]),
    )
    s = score_examples([eg], ["parser"])
    # heads correct on tokens 0,1 -> UAS 2/3; label also correct only on 1
    assert s["dep_uas"] == pytest.approx(2 / 3)
    assert s["dep_las"] == pytest.approx(1 / 3)


def test_ner_prf_math():
    eg = _example(
        ["a", "b", "c", "d"],
        dict(ents=["B-ORG", "L-ORG", "U-PER", "O"]),
        dict(ents=["B-ORG", "L-ORG", "O", "U-PER"]),
    )
    s = score_examples([eg], ["ner"])
    # gold spans: (0,2,ORG),(2,3,PER); pred: (0,2,ORG),(3,4,PER) -> tp=1,fp=1,fn=1
    assert s["ents_p"] == pytest.approx(0.5)
    assert s["ents_r"] == pytest.approx(0.5)
    assert s["ents_f"] == pytest.approx(0.5)


def test_ents_to_spans_malformed_sequences_drop_open_span():
    # L without B, I without B, label switch mid-span
    assert _ents_to_spans(["L-ORG"]) == set()
    assert _ents_to_spans(["I-ORG", "L-ORG"]) == set()
    assert _ents_to_spans(["B-ORG", "I-PER", "L-ORG"]) == set()
    assert _ents_to_spans(["B-ORG", "L-ORG"]) == {(0, 2, "ORG")}


def test_weighted_score_skips_missing_and_none():
    s = weighted_score({"tag_acc": 0.5, "speed": 100.0, "x": None},
                       {"tag_acc": 2.0, "missing": 1.0, "x": 1.0})
    assert s == pytest.approx(1.0)


def test_uas_las_excludes_punctuation():
    # token 2's gold dep is punct: excluded from UAS/LAS entirely
    eg = _example(
        ["a", "b", "."],
        dict(heads=[1, -1, 1], deps=["d1", "ROOT", "punct"]),
        dict(heads=[1, -1, 0], deps=["d1", "ROOT", "punct"]),
    )
    s = score_examples([eg], ["parser"])
    assert s["dep_uas"] == pytest.approx(1.0)
    assert s["dep_las"] == pytest.approx(1.0)
