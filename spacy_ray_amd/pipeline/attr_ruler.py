"""AttributeRuler: pattern-driven token attribute overrides (spaCy's
`attribute_ruler` — in real en_core_web_* pipelines it fixes tags/lemmas
for exceptional tokens after the statistical components run).

Rules follow spaCy's shape: a token-spec pattern (the entity_ruler
matcher subset: ORTH/TEXT/LOWER/IS_DIGIT/IS_ALPHA), an `attrs` dict to
assign, and an `index` selecting which matched token receives them
(default 0; negative indexes from the end of the match).  Supported
attrs: TAG, LEMMA, MORPH.  Rules serialize via cfg.json."""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

from .pipes import TrainablePipe
from .ruler import _tok_match, _validate

_ATTRS = ("TAG", "LEMMA", "MORPH")


class AttributeRulerPipe(TrainablePipe):
    name = "attribute_ruler"
    trainable = False
    listens_to = None

    def __init__(self, name: str = "attribute_ruler", model=None,
                 patterns: Optional[List[Dict]] = None) -> None:
        super().__init__()
        self.name = name
        self.embedded_spec = None
        self.patterns: List[Dict] = []
        if patterns:
            self.add_patterns(patterns)

    def add_patterns(self, patterns: Sequence[Dict]) -> None:
        for entry in patterns:
            pats = entry.get("patterns")
            attrs = entry.get("attrs")
            if not pats or not isinstance(attrs, dict) or not attrs:
                raise ValueError(
                    f"attribute_ruler: each rule needs 'patterns' (list of "
                    f"token-spec lists) and a non-empty 'attrs' dict: {entry!r}")
            for key in attrs:
                if key not in _ATTRS:
                    raise ValueError(
                        f"attribute_ruler: unsupported attr {key!r} "
                        f"(supported: {', '.join(_ATTRS)})")
            for pat in pats:
                _validate(pat if not isinstance(pat, str) else pat)
            self.patterns.append({
                "patterns": [list(p) if not isinstance(p, str) else p
                             for p in pats],
                "attrs": dict(attrs),
                "index": int(entry.get("index", 0)),
            })

    # --------------------------------------------------------- pipe protocol
    def initialize(self, examples, device) -> None:
        pass

    def state_cfg(self) -> Dict:
        cfg = dict(self.cfg)
        cfg["patterns"] = self.patterns
        return cfg

    def load_cfg(self, cfg: Dict, device) -> None:
        self.cfg = dict(cfg)
        self.patterns = list(cfg.get("patterns", []))

    def get_loss(self, examples, t2v, batch):  # pragma: no cover - skipped
        raise RuntimeError("attribute_ruler is not trainable")

    def predict_and_set(self, docs, t2v=None, batch=None) -> None:
        for doc in docs:
            self._apply_doc(doc)

    def __call__(self, docs) -> None:
        self.predict_and_set(docs)

    def _apply_doc(self, doc) -> None:
        n = len(doc)
        words = doc.words
        for rule in self.patterns:
            attrs = rule["attrs"]
            index = rule["index"]
            for pat in rule["patterns"]:
                toks = ([{"ORTH": w} for w in pat.split()]
                        if isinstance(pat, str) else pat)
                m = len(toks)
                for i in range(0, n - m + 1):
                    if all(_tok_match(toks[k], words[i + k]) for k in range(m)):
                        j = i + (index if index >= 0 else m + index)
                        if not (i <= j < i + m):
                            continue
                        self._set_attrs(doc, j, attrs)

    @staticmethod
    def _set_attrs(doc, j: int, attrs: Dict) -> None:
        n = len(doc)
        if "TAG" in attrs:
            if doc.tags is None:
                doc.tags = [""] * n
            doc.tags[j] = attrs["TAG"]
        if "MORPH" in attrs:
            if doc.morphs is None:
                doc.morphs = [""] * n
            doc.morphs[j] = attrs["MORPH"]
        if "LEMMA" in attrs:
            if getattr(doc, "lemmas", None) is None:
                doc.lemmas = [""] * n
            doc.lemmas[j] = attrs["LEMMA"]
