"""Token alignment between two tokenizations of the same text.

Behavioral contract of spaCy's `spacy.training.align` (SURVEY.md §2.2 N9;
upstream spacy/training/align.pyx — re-implemented from the char-offset
formulation): both token sequences must spell the same character string
once whitespace is ignored; each token maps to the tokens of the other
side whose character spans overlap.  `Example.from_docs` uses this to
project reference annotations (tags, BILUO entities, sentence starts)
onto a differently-tokenized predicted Doc — tokens without a clean
alignment get spaCy's missing marker ('-' / None).  Dependency heads are
projected only for 1:1-aligned tokens whose head is also 1:1-aligned.
"""
from __future__ import annotations

from typing import List, Tuple

import numpy as np


class Alignment:
    __slots__ = ("a2b", "b2a")

    def __init__(self, a2b: List[List[int]], b2a: List[List[int]]) -> None:
        self.a2b = a2b
        self.b2a = b2a


def _char_spans(words) -> Tuple[str, List[Tuple[int, int]]]:
    text = []
    spans = []
    pos = 0
    for w in words:
        w = "".join(w.split())  # tokens never contain meaningful whitespace
        text.append(w)
        spans.append((pos, pos + len(w)))
        pos += len(w)
    return "".join(text), spans


def get_alignment(words_a, words_b) -> Alignment:
    text_a, spans_a = _char_spans(words_a)
    text_b, spans_b = _char_spans(words_b)
    if text_a != text_b:
        raise ValueError(
            f"cannot align tokenizations of different texts: "
            f"{text_a[:60]!r} vs {text_b[:60]!r} (spaCy E949 contract)"
        )
    a2b: List[List[int]] = [[] for _ in words_a]
    b2a: List[List[int]] = [[] for _ in words_b]
    j = 0
    for i, (sa, ea) in enumerate(spans_a):
        if sa == ea:
            continue
        while j < len(spans_b) and spans_b[j][1] <= sa:
            j += 1
        k = j
        while k < len(spans_b) and spans_b[k][0] < ea:
            if spans_b[k][1] > spans_b[k][0]:  # skip empty tokens
                a2b[i].append(k)
                b2a[k].append(i)
            k += 1
    return Alignment(a2b, b2a)


def project_reference(predicted, reference):
    """Project reference annotations onto the predicted tokenization
    (IN PLACE on a copy of nothing — returns the arrays; caller assigns).
    Returns (tags, ents, sent_starts, heads, deps) — any of which may be
    None when the reference lacks that annotation."""
    align = get_alignment(predicted.words, reference.words)
    n = len(predicted.words)
    one2one = [len(bs) == 1 and len(align.b2a[bs[0]]) == 1
               for bs in align.a2b]

    tags = None
    if reference.tags is not None:
        tags = [reference.tags[align.a2b[i][0]] if one2one[i] else "-"
                for i in range(n)]

    sents = None
    if reference.sent_starts is not None:
        sents = np.zeros(n, dtype=np.int32)
        for i in range(n):
            for b in align.a2b[i]:
                if reference.sent_starts[b]:
                    sents[i] = 1
                    break

    ents = None
    if reference.ents is not None:
        # gold entity spans in reference token space -> predicted tokens
        # covering exactly those chars; partially-covered pred tokens make
        # the whole span unalignable -> '-' (spaCy's behavior)
        ents = ["O"] * n
        # mark tokens overlapping any missing ('-') reference token
        for b, tag in enumerate(reference.ents):
            if tag in (None, "-"):
                for a in align.b2a[b]:
                    ents[a] = "-"
        spans = _biluo_spans(reference.ents)
        for (bs, be, label) in spans:
            a_toks = sorted({a for b in range(bs, be) for a in align.b2a[b]})
            if not a_toks:
                continue
            # clean iff the predicted tokens cover exactly the same ref span
            covered = sorted({b for a in a_toks for b in align.a2b[a]})
            if covered == list(range(bs, be)):
                if len(a_toks) == 1:
                    ents[a_toks[0]] = f"U-{label}"
                else:
                    ents[a_toks[0]] = f"B-{label}"
                    for a in a_toks[1:-1]:
                        ents[a] = f"I-{label}"
                    ents[a_toks[-1]] = f"L-{label}"
            else:
                for a in a_toks:
                    ents[a] = "-"

    heads = deps = None
    if reference.heads is not None:
        heads = np.full(n, -1, dtype=np.int32)
        deps = [""] * n
        ref_deps = reference.deps or [""] * len(reference.words)
        for i in range(n):
            if not one2one[i]:
                continue
            b = align.a2b[i][0]
            h = int(reference.heads[b])
            if h < 0:
                heads[i] = -1
                deps[i] = ref_deps[b]
            elif len(align.b2a[h]) == 1:
                heads[i] = align.b2a[h][0]
                deps[i] = ref_deps[b]
    return tags, ents, sents, heads, deps


def _biluo_spans(ents) -> List[Tuple[int, int, str]]:
    spans = []
    start, label = None, None
    for i, tag in enumerate(ents or []):
        if tag in (None, "O", "-", ""):
            start, label = None, None
            continue
        kind, _, lab = tag.partition("-")
        if kind == "U":
            spans.append((i, i + 1, lab))
            start, label = None, None
        elif kind == "B":
            start, label = i, lab
        elif kind == "L" and start is not None and lab == label:
            spans.append((start, i + 1, lab))
            start, label = None, None
        elif kind != "I" or lab != label:
            start, label = None, None
    return spans
