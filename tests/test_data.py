"""DocBin round-trip, Corpus.v1 reader, batchers, attrs."""
import numpy as np
import pytest

from spacy_ray_amd.config.registry import registry
from spacy_ray_amd.data.batcher import (configure_minibatch,
                                        configure_minibatch_by_words)
from spacy_ray_amd.data.docbin import DocBin
from spacy_ray_amd.vocab.attrs import word_shape
from spacy_ray_amd.vocab.doc import Doc, Vocab


def test_word_shape():
    assert word_shape("Apple") == "Xxxxx"
    assert word_shape("ABC123") == "XXXddd"
    assert word_shape("aaaaaaaa") == "xxxx"
    assert word_shape("don't") == "xxx'x"


def test_attr_hashes_shape(vocab):
    d = Doc(vocab, ["Hello", "world", "!"])
    assert d.attr_hashes.shape == (3, 4)
    assert d.attr_hashes.dtype == np.uint64
    # same word -> same attr hashes
    d2 = Doc(vocab, ["Hello"])
    assert (d.attr_hashes[0] == d2.attr_hashes[0]).all()


def test_docbin_roundtrip(tmp_path, vocab):
    docs = [
        Doc(vocab, ["a", "b", "c"], tags=["T1", "T2", "T1"],
            heads=[1, -1, 1], deps=["dep0", "ROOT", "dep1"],
            ents=["O", "U-ORG", "O"]),
        Doc(vocab, ["x", "y"]),
    ]
    db = DocBin(docs)
    p = tmp_path / "data.spacy"
    db.to_disk(p)
    db2 = DocBin.from_disk(p, vocab)
    assert len(db2) == 2
    d = db2.docs[0]
    assert d.words == ["a", "b", "c"]
    assert d.tags == ["T1", "T2", "T1"]
    assert d.heads.tolist() == [1, -1, 1]
    assert d.ents == ["O", "U-ORG", "O"]
    assert db2.docs[1].tags is None


def test_corpus_v1_reader(tmp_path, vocab):
    registry.ensure_populated()
    docs = [Doc(vocab, [f"w{i}" for i in range(n)], tags=["T"] * n)
            for n in (3, 5, 7)]
    DocBin(docs).to_disk(tmp_path / "train.spacy")
    reader = registry.readers.get("spacy.Corpus.v1")(path=str(tmp_path / "train.spacy"),
                                                     max_length=6)

    class FakeNlp:
        pass

    FakeNlp.vocab = vocab
    egs = list(reader(FakeNlp))
    assert len(egs) == 2  # 7-word doc filtered by max_length
    assert egs[0].reference.tags == ["T", "T", "T"]
    assert egs[0].predicted.tags is None  # predicted side unannotated


def test_batch_by_words():
    batcher = configure_minibatch_by_words(size=10, tolerance=0.0)
    items = [[0] * n for n in (4, 4, 4, 9, 20, 2)]
    batches = list(batcher(items))
    # 4+4 fits, 4+9 overflows -> [4,4], [4], [9], [20] oversize own batch, [2]
    sizes = [sum(len(x) for x in b) for b in batches]
    assert all(s <= 10 or len(b) == 1 for s, b in zip(sizes, batches))
    assert sum(len(b) for b in batches) == 6  # nothing dropped


def test_batch_by_words_discard_oversize():
    batcher = configure_minibatch_by_words(size=10, tolerance=0.0, discard_oversize=True)
    items = [[0] * n for n in (4, 20, 4)]
    batches = list(batcher(items))
    assert sum(len(b) for b in batches) == 2


def test_batch_by_sequence():
    batcher = configure_minibatch(size=2)
    batches = list(batcher([1, 2, 3, 4, 5]))
    assert [len(b) for b in batches] == [2, 2, 1]


def test_schedule_compounding():
    registry.ensure_populated()
    sched = registry.schedules.get("compounding.v1")(start=1.0, stop=8.0, compound=2.0)
    vals = [next(sched) for _ in range(5)]
    assert vals == [1.0, 2.0, 4.0, 8.0, 8.0]


def test_docbin_version_guard():
    import msgpack
    import pytest

    from spacy_ray_amd.data.docbin import DocBin
    from spacy_ray_amd.vocab.doc import Vocab

    bad = msgpack.packb({"version": 99, "docs": []}, use_bin_type=True)
    with pytest.raises(ValueError, match="version"):
        DocBin.from_bytes(bad, Vocab())


def test_murmur2_64a_matches_independent_python_impl():
    """spaCy string hash = MurmurHash64A(seed=1); cross-check the C++
    implementation against an independent python transcription."""
    from spacy_ray_amd import _srx_cpu

    def mm64a(data: bytes, seed: int = 1) -> int:
        m = 0xC6A4A7935BD1E995
        r = 47
        M = (1 << 64) - 1
        h = (seed ^ ((len(data) * m) & M)) & M
        n8 = len(data) // 8
        for i in range(n8):
            k = int.from_bytes(data[i * 8:(i + 1) * 8], "little")
            k = (k * m) & M
            k ^= k >> r
            k = (k * m) & M
            h ^= k
            h = (h * m) & M
        tail = data[n8 * 8:]
        for j in range(len(tail) - 1, -1, -1):
            h ^= tail[j] << (8 * j)
        if tail:
            h = (h * m) & M
        h ^= h >> r
        h = (h * m) & M
        h ^= h >> r
        return h

    for s in ["", "a", "dog", "ORG", "antidisestablishmentarianism", "héllo"]:
        assert int(_srx_cpu.spacy_hash_string(s)) == mm64a(s.encode("utf8")), s


def test_spacy_docbin_format_structure_and_roundtrip():
    """The written bytes follow the real `.spacy` layout (zlib + msgpack with
    spaCy's keys; tokens = uint64 [T, n_attrs]; strings referenced by
    MurmurHash64A) and round-trip through our reader."""
    import zlib

    import msgpack
    import numpy as np

    from spacy_ray_amd.data.docbin import DocBin, SPACY_ATTRS
    from spacy_ray_amd import _srx_cpu
    from spacy_ray_amd.vocab.doc import Doc, Vocab

    v = Vocab()
    d1 = Doc(v, ["Apple", "buys", "a", "startup"],
             tags=["NNP", "VBZ", "DT", "NN"],
             heads=[1, -1, 3, 1],
             deps=["nsubj", "ROOT", "det", "dobj"],
             ents=["U-ORG", "O", "O", "O"])
    d2 = Doc(v, ["a", "b", "c"], ents=["B-PER", "L-PER", "-"])
    data = DocBin([d1, d2]).to_bytes()
    msg = msgpack.unpackb(zlib.decompress(data), raw=False)
    assert msg["version"] == "0.1"
    assert msg["attrs"] == SPACY_ATTRS
    tokens = np.frombuffer(msg["tokens"], dtype=np.uint64).reshape(-1, len(SPACY_ATTRS))
    assert tokens.shape[0] == 7
    assert np.frombuffer(msg["lengths"], dtype=np.int32).tolist() == [4, 3]
    # ORTH of token 0 is the spaCy hash of "Apple"
    assert int(tokens[0, 0]) == int(_srx_cpu.spacy_hash_string("Apple"))
    # HEAD column is RELATIVE (head - i), two's complement
    rel = tokens[:, SPACY_ATTRS.index("HEAD")].view(np.int64)
    assert rel[:4].tolist() == [1, 0, 1, -2]
    # ENT_IOB spaCy codes: U->3(B), O->2, B->3, L->1(I), '-'->0
    iob = tokens[:, SPACY_ATTRS.index("ENT_IOB")].tolist()
    assert iob == [3, 2, 2, 2, 3, 1, 0]
    # round-trip
    v2 = Vocab()
    out = DocBin.from_bytes(data, v2)
    r1, r2 = out.docs
    assert r1.words == d1.words and r1.tags == d1.tags
    assert r1.heads.tolist() == [1, -1, 3, 1]
    assert r1.deps == d1.deps and r1.ents == d1.ents
    assert r2.ents == ["B-PER", "L-PER", "-"]


def test_legacy_native_docbin_still_readable():
    from spacy_ray_amd.data.docbin import DocBin
    from spacy_ray_amd.vocab.doc import Doc, Vocab

    v = Vocab()
    bin0 = DocBin([Doc(v, ["x", "y"], tags=["A", "B"])])
    legacy = bin0.to_native_bytes()
    out = DocBin.from_bytes(legacy, Vocab())
    assert out.docs[0].words == ["x", "y"]
    assert out.docs[0].tags == ["A", "B"]


def test_docbin_native_schema_carries_new_doc_fields():
    """The native DocBin schema round-trips cats/morphs/lemmas/spans/
    sent_starts (the `.spacy` wire format stays the documented
    SPACY_ATTRS subset)."""
    from spacy_ray_amd.data.docbin import DocBin
    from spacy_ray_amd.vocab.doc import Doc, Vocab

    v = Vocab()
    d = Doc(v, ["cats", "ran"], tags=["NNS", "VBD"], cats={"X": 1.0},
            morphs=["Number=Plur", ""], lemmas=["cat", "run"],
            spans={"sc": [(0, 2, "S")]}, sent_starts=[1, 0])
    db = DocBin([d])
    d2 = DocBin._from_native_bytes(db.to_native_bytes(), v).docs[0]
    assert d2.cats == {"X": 1.0}
    assert d2.morphs == ["Number=Plur", ""]
    assert d2.lemmas == ["cat", "run"]
    assert d2.spans == {"sc": [(0, 2, "S")]}
    assert list(d2.sent_starts) == [1, 0]
    d3 = DocBin.from_bytes(db.to_bytes(), v).docs[0]
    assert d3.tags == ["NNS", "VBD"]
