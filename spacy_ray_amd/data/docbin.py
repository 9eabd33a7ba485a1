"""DocBin: serialized collections of annotated Docs (msgpack).

Fills the role of spaCy's ``.spacy`` DocBin files with a
msgpack schema of this engine's own Doc dict layout.  Self-consistent
round-trip (write == read) is what the corpus/checkpoint paths need; the
format is versioned for forward compatibility.
"""
from __future__ import annotations

from pathlib import Path
from typing import Iterable, Iterator, List, Union

import msgpack

from spacy_ray_amd.vocab.doc import Doc, Vocab

FORMAT_VERSION = 1


class DocBin:
    def __init__(self, docs: Iterable[Doc] = ()) -> None:
        self.docs: List[Doc] = list(docs)

    def add(self, doc: Doc) -> None:
        self.docs.append(doc)

    def __len__(self) -> int:
        return len(self.docs)

    def to_bytes(self) -> bytes:
        payload = {
            "version": FORMAT_VERSION,
            "docs": [d.to_dict() for d in self.docs],
        }
        return msgpack.packb(payload, use_bin_type=True)

    @classmethod
    def from_bytes(cls, data: bytes, vocab: Vocab) -> "DocBin":
        payload = msgpack.unpackb(data, raw=False)
        if payload.get("version") != FORMAT_VERSION:
            raise ValueError(f"unsupported DocBin version: {payload.get('version')}")
        out = cls()
        for dd in payload["docs"]:
            out.add(Doc.from_dict(vocab, dd))
        return out

    def to_disk(self, path: Union[str, Path]) -> None:
        Path(path).write_bytes(self.to_bytes())

    @classmethod
    def from_disk(cls, path: Union[str, Path], vocab: Vocab) -> "DocBin":
        return cls.from_bytes(Path(path).read_bytes(), vocab)

    def get_docs(self, vocab: Vocab) -> Iterator[Doc]:
        yield from self.docs
