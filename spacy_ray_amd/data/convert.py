"""Corpus converters: CoNLL-U and CoNLL-style IOB -> DocBin (.spacy).

The reference ships `bin/get-data.sh` (fetches a small NER dataset for
manual runs) and otherwise leans on spaCy's `convert` CLI for real data.
There is no network here, so the converter IS the data story: users bring
CoNLL-U treebanks / IOB NER files and convert them to this engine's
DocBin for `spacy.Corpus.v1` readers.

CoNLL-U (https://universaldependencies.org/format.html):
  10 tab columns: ID FORM LEMMA UPOS XPOS FEATS HEAD DEPREL DEPS MISC;
  sentences separated by blank lines, `#` comment lines; multiword-token
  ranges (`1-2`) and empty nodes (`1.1`) are skipped.  HEAD is 1-based
  (0 = root) -> our Doc convention: absolute token index, -1 for root.

IOB (CoNLL-02/03 style):
  token [pos [chunk]] tag  per line, blank line between sentences; the
  trailing column holds IOB1/IOB2 entity tags which are normalized to the
  BILUO scheme our NER pipe trains on.
"""
from __future__ import annotations

from pathlib import Path
from typing import List, Optional

from spacy_ray_amd.vocab.doc import Doc, Vocab


def _finish_conllu_sentence(vocab, words, tags, heads, deps, morphs):
    return Doc(vocab, words, tags=tags, heads=heads, deps=deps,
               morphs=morphs)


def read_conllu(text: str, vocab: Optional[Vocab] = None,
                tag_col: str = "upos") -> List[Doc]:
    """Parse CoNLL-U text into Docs (words, tags, heads, deps)."""
    vocab = vocab or Vocab()
    col = {"upos": 3, "xpos": 4}[tag_col]
    docs: List[Doc] = []
    words: List[str] = []
    tags: List[str] = []
    heads: List[int] = []
    deps: List[str] = []
    morphs: List[str] = []
    for raw in text.splitlines():
        line = raw.rstrip("\n")
        if not line.strip():
            if words:
                docs.append(_finish_conllu_sentence(vocab, words, tags, heads,
                                                    deps, morphs))
                words, tags, heads, deps, morphs = [], [], [], [], []
            continue
        if line.startswith("#"):
            continue
        parts = line.split("\t")
        if len(parts) < 8:
            raise ValueError(f"malformed CoNLL-U line: {line!r}")
        tok_id = parts[0]
        if "-" in tok_id or "." in tok_id:  # multiword range / empty node
            continue
        words.append(parts[1])
        tags.append(parts[col] if parts[col] != "_" else "X")
        head = int(parts[6]) if parts[6] != "_" else 0
        heads.append(head - 1)  # 1-based with 0=root -> index with -1=root
        deps.append(parts[7] if parts[7] != "_" else "dep")
        morphs.append(parts[5] if parts[5] != "_" else "")
    if words:
        docs.append(_finish_conllu_sentence(vocab, words, tags, heads, deps,
                                            morphs))
    return docs


def iob_to_biluo(tags: List[str]) -> List[str]:
    """IOB1/IOB2 entity tags -> BILUO (the scheme the NER pipe trains on)."""
    out: List[str] = []
    n = len(tags)
    for i, tag in enumerate(tags):
        if tag == "O" or not tag:
            out.append("O")
            continue
        kind, _, label = tag.partition("-")
        if not label:  # bare "B"/"I" — treat as O rather than guess
            out.append("O")
            continue
        nxt = tags[i + 1] if i + 1 < n else "O"
        nxt_kind, _, nxt_label = nxt.partition("-")
        continues = nxt_kind == "I" and nxt_label == label
        prev = out[-1] if out else "O"
        open_span = prev.startswith(("B-", "I-")) and prev[2:] == label
        if kind == "B" or (kind == "I" and not open_span):
            out.append(("B-" if continues else "U-") + label)
        else:  # I continuing an open span
            out.append(("I-" if continues else "L-") + label)
    return out


def read_iob(text: str, vocab: Optional[Vocab] = None,
             sep: Optional[str] = None) -> List[Doc]:
    """Parse CoNLL-02/03-style IOB text into Docs (words + BILUO ents;
    a POS column, when present, fills tags)."""
    vocab = vocab or Vocab()
    docs: List[Doc] = []
    words: List[str] = []
    tags: List[str] = []
    ents: List[str] = []

    def finish():
        if words:
            docs.append(Doc(vocab, list(words), tags=list(tags) if any(tags) else None,
                            ents=iob_to_biluo(ents)))
            words.clear(); tags.clear(); ents.clear()

    for raw in text.splitlines():
        line = raw.strip()
        if not line or line.startswith("-DOCSTART-"):
            finish()
            continue
        parts = line.split(sep)
        words.append(parts[0])
        tags.append(parts[1] if len(parts) > 2 else "")
        ents.append(parts[-1] if len(parts) > 1 else "O")
    finish()
    return docs


def convert_file(input_path, output_path, *, fmt: Optional[str] = None,
                 tag_col: str = "upos") -> int:
    """Convert one file to a DocBin `.spacy`; returns the doc count.
    fmt: "conllu" | "iob" | None (inferred from the extension)."""
    from spacy_ray_amd.data.docbin import DocBin

    input_path = Path(input_path)
    if fmt is None:
        suffix = input_path.suffix.lower().lstrip(".")
        fmt = {"conllu": "conllu", "conll": "conllu", "iob": "iob"}.get(suffix)
        if fmt is None:
            raise ValueError(
                f"cannot infer format from {input_path.name!r}; pass --format")
    text = input_path.read_text(encoding="utf-8")
    docs = read_conllu(text, tag_col=tag_col) if fmt == "conllu" else read_iob(text)
    DocBin(docs).to_disk(output_path)
    return len(docs)
