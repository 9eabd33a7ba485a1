"""EntityRuler: rule-based entity annotation (spaCy's `entity_ruler`).

The reference is a generic spaCy trainer, and real spaCy pipelines mix
trainable components with rule-based ones; this covers the most common of
those.  Patterns follow spaCy's shape:

    {"label": "ORG", "pattern": "Apple"}                       # phrase
    {"label": "GPE", "pattern": [{"LOWER": "san"},
                                 {"LOWER": "francisco"}]}      # token specs

Supported token-spec keys: ORTH/TEXT (exact), LOWER (case-insensitive),
IS_DIGIT, IS_ALPHA (booleans).  Unsupported keys raise at add time (loud
failure beats silently never matching).  Matching is longest-first,
left-to-right, non-overlapping; existing entity tokens are preserved
unless ``overwrite_ents``.  Patterns serialize through the component's
cfg.json (no tensors), so checkpoints round-trip like any other pipe.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

from .pipes import TrainablePipe

_SUPPORTED = ("ORTH", "TEXT", "LOWER", "IS_DIGIT", "IS_ALPHA")


def _validate(pattern) -> None:
    if isinstance(pattern, str):
        if not pattern.strip():
            raise ValueError("entity_ruler: empty phrase pattern")
        return
    if not isinstance(pattern, (list, tuple)) or not pattern:
        raise ValueError(f"entity_ruler: pattern must be a string or a "
                         f"non-empty token-spec list, got {pattern!r}")
    for spec in pattern:
        if not isinstance(spec, dict) or not spec:
            raise ValueError(f"entity_ruler: bad token spec {spec!r}")
        for key in spec:
            if key not in _SUPPORTED:
                raise ValueError(
                    f"entity_ruler: unsupported token-spec key {key!r} "
                    f"(supported: {', '.join(_SUPPORTED)})")


def _tok_match(spec: Dict, word: str) -> bool:
    for key, val in spec.items():
        if key in ("ORTH", "TEXT"):
            if word != val:
                return False
        elif key == "LOWER":
            if word.lower() != str(val).lower():
                return False
        elif key == "IS_DIGIT":
            if word.isdigit() != bool(val):
                return False
        elif key == "IS_ALPHA":
            if word.isalpha() != bool(val):
                return False
    return True


class EntityRulerPipe(TrainablePipe):
    name = "entity_ruler"
    trainable = False
    listens_to = None  # rule-based: no encoder input

    def __init__(self, name: str = "entity_ruler", model=None,
                 overwrite_ents: bool = False,
                 patterns: Optional[List[Dict]] = None) -> None:
        super().__init__()
        self.name = name
        self.embedded_spec = None
        self.overwrite_ents = bool(overwrite_ents)
        self.patterns: List[Dict] = []
        self._compiled: List = []
        if patterns:
            self.add_patterns(patterns)

    # ------------------------------------------------------------ patterns
    def add_patterns(self, patterns: Sequence[Dict]) -> None:
        for entry in patterns:
            label = entry.get("label")
            if not label:
                raise ValueError(f"entity_ruler: pattern without label: {entry!r}")
            _validate(entry.get("pattern"))
            self.patterns.append({"label": label, "pattern": entry["pattern"]})
        self._compile()

    def _compile(self) -> None:
        # token-spec form, longest first so the scan is longest-match
        self._compiled = []
        for entry in self.patterns:
            pat = entry["pattern"]
            if isinstance(pat, str):
                toks = [{"ORTH": w} for w in pat.split()]
            else:
                toks = list(pat)
            self._compiled.append((entry["label"], toks))
        self._compiled.sort(key=lambda lp: -len(lp[1]))

    @property
    def labels(self) -> List[str]:
        return sorted({e["label"] for e in self.patterns})

    # --------------------------------------------------------- pipe protocol
    def initialize(self, examples, device) -> None:
        self._compile()

    def state_cfg(self) -> Dict:
        cfg = dict(self.cfg)
        cfg["patterns"] = self.patterns
        cfg["overwrite_ents"] = self.overwrite_ents
        return cfg

    def load_cfg(self, cfg: Dict, device) -> None:
        self.cfg = dict(cfg)
        self.overwrite_ents = bool(cfg.get("overwrite_ents", False))
        self.patterns = list(cfg.get("patterns", []))
        self._compile()

    def get_loss(self, examples, t2v, batch):  # pragma: no cover - skipped
        raise RuntimeError("entity_ruler is not trainable")

    def predict_and_set(self, docs, t2v=None, batch=None) -> None:
        for doc in docs:
            self._annotate_doc(doc)

    def __call__(self, docs) -> None:
        self.predict_and_set(docs)

    def _annotate_doc(self, doc) -> None:
        n = len(doc)
        ents = list(doc.ents) if doc.ents else ["O"] * n
        words = doc.words
        occupied = [e not in ("O", "-", "", None) for e in ents]
        i = 0
        while i < n:
            if occupied[i] and not self.overwrite_ents:
                i += 1
                continue
            matched = False
            for label, toks in self._compiled:
                m = len(toks)
                if i + m > n:
                    continue
                span_free = self.overwrite_ents or not any(
                    occupied[i:i + m])
                if not span_free:
                    continue
                if all(_tok_match(toks[k], words[i + k]) for k in range(m)):
                    if m == 1:
                        ents[i] = f"U-{label}"
                    else:
                        ents[i] = f"B-{label}"
                        for k in range(1, m - 1):
                            ents[i + k] = f"I-{label}"
                        ents[i + m - 1] = f"L-{label}"
                    for k in range(m):
                        occupied[i + k] = True
                    i += m
                    matched = True
                    break
            if not matched:
                i += 1
        doc.ents = ents
