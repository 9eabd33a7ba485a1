"""Token-pattern Matcher (the public face of spaCy's `Matcher` /
`PhraseMatcher`): the same token-spec dialect the entity_ruler /
attribute_ruler components use (ORTH/TEXT exact, LOWER case-insensitive,
IS_DIGIT, IS_ALPHA), exposed as a standalone API:

    from spacy_ray_amd.vocab.matcher import Matcher, PhraseMatcher

    m = Matcher()
    m.add("ORG", [[{"ORTH": "Acme"}, {"ORTH": "Corp"}]])
    matches = m(doc)           # [(key, start, end), ...]

Matches are returned for every rule at every position (overlaps
included), sorted by (start, end) — callers filter (the rulers do
longest-first non-overlapping selection themselves)."""
from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

from spacy_ray_amd.pipeline.ruler import _tok_match, _validate


class Matcher:
    def __init__(self) -> None:
        self._rules: List[Tuple[str, List[Dict]]] = []

    def add(self, key: str, patterns: Sequence[Sequence[Dict]]) -> None:
        for pat in patterns:
            _validate(list(pat))
            self._rules.append((key, list(pat)))

    def __len__(self) -> int:
        return len(self._rules)

    def __call__(self, doc) -> List[Tuple[str, int, int]]:
        words = doc.words if hasattr(doc, "words") else list(doc)
        n = len(words)
        out: List[Tuple[str, int, int]] = []
        for key, toks in self._rules:
            m = len(toks)
            for i in range(0, n - m + 1):
                if all(_tok_match(toks[k], words[i + k]) for k in range(m)):
                    out.append((key, i, i + m))
        out.sort(key=lambda t: (t[1], t[2]))
        return out


class PhraseMatcher(Matcher):
    """Phrase variant: patterns are token lists (or space-joined strings)
    matched by exact ORTH."""

    def add(self, key: str, phrases: Sequence) -> None:  # type: ignore[override]
        for phrase in phrases:
            words = (phrase.split() if isinstance(phrase, str)
                     else [w if isinstance(w, str) else w for w in phrase])
            if not words:
                raise ValueError("PhraseMatcher: empty phrase")
            self._rules.append((key, [{"ORTH": w} for w in words]))
