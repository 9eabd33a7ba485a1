"""Lemmatizer (spaCy's `lemmatizer` factory): rule and lookup modes.

spaCy's rule lemmatizer drives POS-indexed suffix rules + exception
tables shipped in `spacy-lookups-data`; there is no network here, so the
rule mode carries a compact built-in English rule set (regular noun /
verb / adjective suffix stripping + the common irregulars) and the
lookup mode takes a user table through the component config:

    [components.lemmatizer]
    factory = "lemmatizer"
    mode = "lookup"                       # or "rule" (default)
    lookups = {"went": "go", ...}         # lookup mode table

Lemmas land in `Doc.lemmas`; tokens with no matching rule keep their
lower-cased form (spaCy's fallback).  Runs after the tagger (rule mode
reads PTB-style tags when present).  Scored as lemma_acc."""
from __future__ import annotations

from typing import Dict, Optional

from .pipes import TrainablePipe

# the common English irregulars (verbs, nouns, comparatives)
_EXC = {
    "am": "be", "are": "be", "is": "be", "was": "be", "were": "be",
    "been": "be", "being": "be",
    "has": "have", "had": "have", "having": "have",
    "does": "do", "did": "do", "done": "do", "doing": "do",
    "goes": "go", "went": "go", "gone": "go", "going": "go",
    "said": "say", "says": "say",
    "made": "make", "took": "take", "taken": "take", "came": "come",
    "saw": "see", "seen": "see", "got": "get", "gotten": "get",
    "gave": "give", "given": "give", "found": "find", "knew": "know",
    "known": "know", "thought": "think", "told": "tell", "left": "leave",
    "felt": "feel", "kept": "keep", "held": "hold", "brought": "bring",
    "wrote": "write", "written": "write", "stood": "stand", "met": "meet",
    "ran": "run", "paid": "pay", "sat": "sit", "spoke": "speak",
    "spoken": "speak", "lay": "lie", "led": "lead", "grew": "grow",
    "grown": "grow", "lost": "lose", "fell": "fall", "fallen": "fall",
    "sent": "send", "built": "build", "understood": "understand",
    "drew": "draw", "drawn": "draw", "broke": "break", "broken": "break",
    "spent": "spend", "cut": "cut", "put": "put",
    "children": "child", "men": "man", "women": "woman", "people": "person",
    "mice": "mouse", "feet": "foot", "teeth": "tooth", "geese": "goose",
    "lives": "life", "wives": "wife", "knives": "knife", "leaves": "leaf",
    "better": "good", "best": "good", "worse": "bad", "worst": "bad",
    "further": "far", "furthest": "far",
}

_VOWELS = set("aeiou")


def _rule_lemma(word: str, tag: str) -> str:
    low = word.lower()
    if low in _EXC:
        return _EXC[low]
    t = tag or ""
    if t.startswith("NN"):  # nouns: plural stripping
        if t in ("NNS", "NNPS") or t == "":
            if low.endswith("ies") and len(low) > 4:
                return low[:-3] + "y"
            if low.endswith(("ches", "shes", "sses", "xes", "zes")):
                return low[:-2]
            if low.endswith("s") and not low.endswith(("ss", "us", "is")) \
                    and len(low) > 3:
                return low[:-1]
        return low
    if t.startswith("VB"):  # verbs
        if low.endswith("ies") and len(low) > 4:
            return low[:-3] + "y"
        if low.endswith(("ches", "shes", "sses", "xes", "zes")):
            return low[:-2]
        if low.endswith("s") and not low.endswith("ss") and len(low) > 3:
            return low[:-1]
        if low.endswith("ying") and len(low) > 5:
            return low[:-4] + "y"
        if low.endswith("ing") and len(low) > 5:
            stem = low[:-3]
            if len(stem) > 2 and stem[-1] == stem[-2] and stem[-1] not in _VOWELS:
                return stem[:-1]  # running -> run
            if len(stem) > 2 and stem[-1] not in _VOWELS and stem[-2] in _VOWELS \
                    and stem[-3] not in _VOWELS:
                return stem  # CVC: sitting handled above; walking -> walk
            return stem + "e" if stem[-1] in "uvcs" else stem
        if low.endswith("ied") and len(low) > 4:
            return low[:-3] + "y"
        if low.endswith("ed") and len(low) > 4:
            stem = low[:-2]
            if len(stem) > 2 and stem[-1] == stem[-2] and stem[-1] not in _VOWELS:
                return stem[:-1]  # stopped -> stop
            if stem.endswith(("at", "iz", "is", "u", "v")):
                return stem + "e"
            return stem
        return low
    if t in ("JJR", "JJS", "RBR", "RBS"):  # comparatives/superlatives
        if low.endswith("ier"):
            return low[:-3] + "y"
        if low.endswith("iest"):
            return low[:-4] + "y"
        if low.endswith("er") and len(low) > 4:
            return low[:-2]
        if low.endswith("est") and len(low) > 5:
            return low[:-3]
        return low
    return low


class LemmatizerPipe(TrainablePipe):
    name = "lemmatizer"
    trainable = False
    listens_to = None

    def __init__(self, name: str = "lemmatizer", model=None,
                 mode: str = "rule",
                 lookups: Optional[Dict[str, str]] = None,
                 overwrite: bool = False) -> None:
        super().__init__()
        if mode not in ("rule", "lookup"):
            raise ValueError(f"lemmatizer: mode must be rule|lookup, got {mode!r}")
        self.name = name
        self.embedded_spec = None
        self.mode = mode
        self.lookups = dict(lookups or {})
        self.overwrite = bool(overwrite)

    def initialize(self, examples, device) -> None:
        pass

    def state_cfg(self) -> Dict:
        cfg = dict(self.cfg)
        cfg.update(mode=self.mode, lookups=self.lookups,
                   overwrite=self.overwrite)
        return cfg

    def load_cfg(self, cfg: Dict, device) -> None:
        self.cfg = dict(cfg)
        self.mode = cfg.get("mode", "rule")
        self.lookups = dict(cfg.get("lookups", {}))
        self.overwrite = bool(cfg.get("overwrite", False))

    def get_loss(self, examples, t2v, batch):  # pragma: no cover - skipped
        raise RuntimeError("lemmatizer is not trainable")

    def predict_and_set(self, docs, t2v=None, batch=None) -> None:
        for doc in docs:
            self._lemmatize_doc(doc)

    def __call__(self, docs) -> None:
        self.predict_and_set(docs)

    def _lemmatize_doc(self, doc) -> None:
        n = len(doc)
        if doc.lemmas is None:
            doc.lemmas = [""] * n
        tags = doc.tags or [""] * n
        for i, word in enumerate(doc.words):
            if doc.lemmas[i] and not self.overwrite:
                continue  # e.g. attribute_ruler already set it
            if self.mode == "lookup":
                doc.lemmas[i] = self.lookups.get(word.lower(), word.lower())
            else:
                doc.lemmas[i] = _rule_lemma(word, tags[i])
