"""Char-offset token alignment + annotation projection."""
import numpy as np
import pytest

from spacy_ray_amd.vocab.align import get_alignment, project_reference
from spacy_ray_amd.vocab.doc import Doc, Example, Vocab


def test_alignment_one_to_many_and_many_to_one():
    a = ["i", "listened", "to", "obama", "'s", "podcasts", "."]
    b = ["i", "listened", "to", "obama's", "podcasts."]
    al = get_alignment(a, b)
    assert al.a2b == [[0], [1], [2], [3], [3], [4], [4]]
    assert al.b2a == [[0], [1], [2], [3, 4], [5, 6]]


def test_alignment_mismatched_text_raises():
    with pytest.raises(ValueError):
        get_alignment(["abc"], ["abd"])


def test_project_tags_and_ents():
    v = Vocab()
    ref = Doc(v, ["New", "York", "is", "big", "."],
              tags=["NNP", "NNP", "VBZ", "JJ", "."],
              ents=["B-GPE", "L-GPE", "O", "O", "O"])
    pred = Doc(v, ["New York", "is", "big", "."])
    eg = Example.from_docs(pred, ref)
    # "New York" merged: tag unalignable -> '-'; entity aligns cleanly (the
    # merged token covers exactly the gold span)
    assert eg.reference.tags == ["-", "VBZ", "JJ", "."]
    assert eg.reference.ents == ["U-GPE", "O", "O", "O"]


def test_project_entity_partial_overlap_is_missing():
    v = Vocab()
    ref = Doc(v, ["Fort", "Knox", "gold"], ents=["B-FAC", "L-FAC", "O"])
    pred = Doc(v, ["Fort", "Knoxgold"])  # pred token straddles the boundary
    eg = Example.from_docs(pred, ref)
    assert eg.reference.ents[1] == "-"


def test_project_heads_one_to_one_only():
    v = Vocab()
    ref = Doc(v, ["the", "big", "dog"], heads=[2, 2, -1],
              deps=["det", "amod", "ROOT"])
    pred = Doc(v, ["the", "big", "dog"])
    eg = Example.from_docs(pred, ref)  # same tokenization: direct pair
    assert eg.reference is ref
    pred2 = Doc(v, ["the big", "dog"])
    eg2 = Example.from_docs(pred2, ref)
    # merged token has no 1:1 alignment; "dog" does and its head (-1) maps
    assert eg2.reference.heads.tolist() == [-1, -1]
    assert eg2.reference.deps[1] == "ROOT"


def test_project_sent_starts():
    v = Vocab()
    ref = Doc(v, ["a", "b", "c", "d"], sent_starts=[1, 0, 1, 0])
    pred = Doc(v, ["ab", "cd"])
    eg = Example.from_docs(pred, ref)
    assert eg.reference.sent_starts.tolist() == [1, 1]
