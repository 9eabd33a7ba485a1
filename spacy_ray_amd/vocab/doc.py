"""Doc / Vocab / Example.

MI355X-first re-design of spaCy's Cython Doc/Vocab: a Doc
is a contiguous struct-of-arrays — token texts plus a precomputed
(n_tokens, 4) uint64 attr-hash matrix that ships to the GPU as one int64
tensor.  Annotations (tags / heads / deps / BILUO ents) are plain arrays.
Examples pair a predicted Doc with a reference (gold) Doc, the contract of
spaCy's training Example consumed by nlp.update.
"""
from __future__ import annotations

import re
from typing import Dict, Iterator, List, Optional, Sequence

import numpy as np

from .attrs import extract_attr_hashes
from .strings import StringStore

_TOKEN_RE = re.compile(r"\w+|[^\w\s]", re.UNICODE)


class Vocab:
    def __init__(self, lang: str = "xx") -> None:
        self.lang = lang
        self.strings = StringStore()

    def __repr__(self) -> str:
        return f"Vocab(lang={self.lang!r}, strings={len(self.strings)})"


class Doc:
    __slots__ = (
        "vocab", "words", "spaces", "attr_hashes",
        "tags", "heads", "deps", "ents", "sent_starts", "cats", "morphs",
        "lemmas", "spans", "tensor", "user_data",
    )

    def __init__(
        self,
        vocab: Vocab,
        words: Sequence[str],
        *,
        spaces: Optional[Sequence[bool]] = None,
        tags: Optional[Sequence[str]] = None,
        heads: Optional[Sequence[int]] = None,
        deps: Optional[Sequence[str]] = None,
        ents: Optional[Sequence[str]] = None,  # per-token BILUO strings, e.g. "B-ORG"/"O"
        sent_starts: Optional[Sequence[int]] = None,  # 1 = starts a sentence
        cats: Optional[Dict[str, float]] = None,  # doc-level categories
        morphs: Optional[Sequence[str]] = None,  # UD FEATS strings per token
        spans: Optional[Dict] = None,  # spans groups: key -> [(start, end, label)]
        lemmas: Optional[Sequence[str]] = None,
        attr_hashes: Optional[np.ndarray] = None,  # precomputed (n,4) uint64
    ) -> None:
        self.vocab = vocab
        self.words = list(words)
        self.spaces = list(spaces) if spaces is not None else [True] * len(self.words)
        # corpora with a known lexicon pass precomputed attr hashes (one
        # lookup per token instead of 4 string builds + hashes)
        self.attr_hashes = (attr_hashes if attr_hashes is not None
                            else extract_attr_hashes(self.words))
        self.tags = list(tags) if tags is not None else None
        self.heads = np.asarray(heads, dtype=np.int32) if heads is not None else None
        self.deps = list(deps) if deps is not None else None
        self.ents = list(ents) if ents is not None else None
        self.sent_starts = (np.asarray(sent_starts, dtype=np.int32)
                            if sent_starts is not None else None)
        self.cats: Optional[Dict[str, float]] = (dict(cats) if cats is not None
                                                 else None)
        self.morphs = list(morphs) if morphs is not None else None
        self.spans: Dict[str, list] = (
            {k: [tuple(sp) for sp in v] for k, v in spans.items()}
            if spans else {})
        self.lemmas = list(lemmas) if lemmas is not None else None
        self.tensor: Optional[np.ndarray] = None
        self.user_data: Dict = {}

    def __len__(self) -> int:
        return len(self.words)

    def __iter__(self) -> Iterator[str]:
        return iter(self.words)

    @property
    def text(self) -> str:
        out = []
        for w, sp in zip(self.words, self.spaces):
            out.append(w)
            if sp:
                out.append(" ")
        return "".join(out).rstrip()

    def copy_unannotated(self) -> "Doc":
        # words are identical, so the (n,4) attr-hash array is shared (it is
        # never mutated) — re-hashing every token per epoch/predict call was
        # pure waste on the Example.from_doc path
        return Doc(self.vocab, self.words, spaces=self.spaces,
                   attr_hashes=self.attr_hashes)

    def to_dict(self) -> Dict:
        return {
            "words": self.words,
            "spaces": self.spaces,
            "tags": self.tags,
            "heads": self.heads.tolist() if self.heads is not None else None,
            "deps": self.deps,
            "ents": self.ents,
            "sent_starts": (self.sent_starts.tolist()
                            if self.sent_starts is not None else None),
            "cats": self.cats,
            "morphs": self.morphs,
            "spans": {k: [list(sp) for sp in v]
                      for k, v in self.spans.items()} or None,
            "lemmas": self.lemmas,
        }

    @classmethod
    def from_dict(cls, vocab: Vocab, data: Dict) -> "Doc":
        return cls(
            vocab,
            data["words"],
            spaces=data.get("spaces"),
            tags=data.get("tags"),
            heads=data.get("heads"),
            deps=data.get("deps"),
            ents=data.get("ents"),
            sent_starts=data.get("sent_starts"),
            cats=data.get("cats"),
            morphs=data.get("morphs"),
            spans=data.get("spans"),
            lemmas=data.get("lemmas"),
        )


class Example:
    """predicted + reference Doc pair (contract of spaCy's Example)."""

    __slots__ = ("predicted", "reference")

    def __init__(self, predicted: Doc, reference: Doc) -> None:
        self.predicted = predicted
        self.reference = reference

    @classmethod
    def from_doc(cls, doc: Doc) -> "Example":
        return cls(doc.copy_unannotated(), doc)

    @classmethod
    def from_docs(cls, predicted: Doc, reference: Doc) -> "Example":
        """Pair a predicted Doc with a DIFFERENTLY-tokenized reference: the
        reference annotations are char-aligned and projected onto the
        predicted tokenization (spaCy Example(predicted, reference)
        contract; unalignable tokens get the missing marker).  Same
        tokenization short-circuits to a direct pair."""
        if predicted.words == reference.words:
            return cls(predicted, reference)
        from .align import project_reference

        tags, ents, sents, heads, deps = project_reference(predicted, reference)
        projected = Doc(
            predicted.vocab, predicted.words, spaces=predicted.spaces,
            tags=tags, heads=heads, deps=deps, ents=ents, sent_starts=sents,
            attr_hashes=predicted.attr_hashes,
        )
        return cls(predicted, projected)

    def __len__(self) -> int:
        return len(self.reference)

    @property
    def x(self) -> Doc:
        return self.predicted

    @property
    def y(self) -> Doc:
        return self.reference


def simple_tokenize(vocab: Vocab, text: str) -> Doc:
    """Rule-based fallback tokenizer (word chars / single punct).  Training
    from pre-annotated corpora never calls this (SURVEY.md §2.2 N5 note:
    tokenization is off the hot path)."""
    words = _TOKEN_RE.findall(text)
    return Doc(vocab, words)


def biluo_to_codes(ents: Optional[Sequence[str]], label2id: Dict[str, int]) -> np.ndarray:
    """Per-token BILUO strings -> int codes: 0=O; for type t: 1+4t=B, 2+4t=I,
    3+4t=L, 4+4t=U; -1 = MISSING (spaCy's '-' unannotated marker: the token
    is excluded from supervision, not negative evidence — ADVICE r1).
    Layout shared with ops/csrc/transitions.cpp BiluoBatch."""
    if ents is None:
        return np.zeros(0, dtype=np.int32)
    kinds = {"B": 0, "I": 1, "L": 2, "U": 3}
    out = np.zeros(len(ents), dtype=np.int32)
    for i, tag in enumerate(ents):
        if tag in (None, "-"):
            out[i] = -1  # missing: no supervision for this token
        elif tag in ("O", ""):
            out[i] = 0
        else:
            kind, _, label = tag.partition("-")
            t = label2id.get(label)
            if t is None:
                raise ValueError(
                    f"NER gold label {label!r} is not in the component's label "
                    f"set {sorted(label2id)} — labels are discovered over the "
                    f"full training corpus at init (or pinned via the "
                    f"component's `labels` config); rebuild the label set"
                )
            out[i] = 1 + 4 * t + kinds[kind]
    return out


def biluo_string_table(id2label: List[str]) -> np.ndarray:
    """Action-code -> BILUO-string lookup table (index 0 = 'O'), for
    vectorized decode annotation (the per-token python loop dominated
    serve-path latency at 200k-word batches)."""
    kinds = "BILU"
    table = ["O"]
    for label in id2label:
        for k in kinds:
            table.append(f"{k}-{label}")
    return np.asarray(table, dtype=object)


def codes_to_biluo(codes: np.ndarray, id2label: List[str]) -> List[str]:
    codes = np.asarray(codes)
    table = biluo_string_table(id2label)
    safe = np.where((codes > 0) & (codes < len(table)), codes, 0)
    return table[safe].tolist()
