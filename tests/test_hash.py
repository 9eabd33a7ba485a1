"""MurmurHash3 x64_128 correctness (the HashEmbed row-hash contract)."""
import numpy as np

from spacy_ray_amd import _srx_cpu


def _rotl64(x, r):
    return ((x << r) | (x >> (64 - r))) & (2**64 - 1)


def _fmix64(k):
    k ^= k >> 33
    k = (k * 0xFF51AFD7ED558CCD) % 2**64
    k ^= k >> 33
    k = (k * 0xC4CEB9FE1A85EC53) % 2**64
    k ^= k >> 33
    return k


def _mm_x64_128_u64(key, seed):
    c1, c2 = 0x87C37B91114253D5, 0x4CF5AD432745937F
    h1 = h2 = seed
    k1 = (key * c1) % 2**64
    k1 = _rotl64(k1, 31)
    k1 = (k1 * c2) % 2**64
    h1 ^= k1
    h1 ^= 8
    h2 ^= 8
    h1 = (h1 + h2) % 2**64
    h2 = (h2 + h1) % 2**64
    h1 = _fmix64(h1)
    h2 = _fmix64(h2)
    h1 = (h1 + h2) % 2**64
    h2 = (h2 + h1) % 2**64
    return [h1 & 0xFFFFFFFF, h1 >> 32, h2 & 0xFFFFFFFF, h2 >> 32]


def test_hash4_matches_independent_python_impl():
    rng = np.random.RandomState(0)
    ids = rng.randint(0, 2**63, size=64, dtype=np.int64).view(np.uint64)
    ids[0] = 0
    ids[1] = 2**64 - 1
    for seed in (0, 1, 7, 123456):
        got = _srx_cpu.hash4(ids, seed)
        for i, k in enumerate(ids.tolist()):
            assert got[i].tolist() == _mm_x64_128_u64(k, seed)


def test_hash4_lanes_distinct():
    ids = np.arange(1, 1000, dtype=np.uint64)
    h = _srx_cpu.hash4(ids, 5)
    # all 4 lanes should differ for virtually every key
    same = sum(len(set(row)) < 4 for row in h.tolist())
    assert same == 0


def test_hashembed_rows_in_range_and_stable():
    ids = np.arange(100, dtype=np.uint64) * 2654435761
    rows = _srx_cpu.hashembed_rows(ids, 3, 500)
    assert rows.shape == (100, 4)
    assert rows.min() >= 0 and rows.max() < 500
    rows2 = _srx_cpu.hashembed_rows(ids, 3, 500)
    assert (rows == rows2).all()
    h = _srx_cpu.hash4(ids, 3)
    assert ((h % 500).astype(np.int32) == rows).all()


def test_hash_string_stable():
    h1 = _srx_cpu.hash_string("hello")
    h2 = _srx_cpu.hash_strings(["hello", "world"])
    assert h1 == h2[0]
    assert h2[0] != h2[1]
    assert _srx_cpu.hash_string("") != _srx_cpu.hash_string(" ")


def test_word_shape_contract():
    """spaCy word_shape semantics: case/digit classes, run truncation at 4,
    long-string short-circuit."""
    from spacy_ray_amd.vocab.attrs import word_shape

    assert word_shape("Apple") == "Xxxxx"
    assert word_shape("USA") == "XXX"
    assert word_shape("C3PO") == "XdXX"
    assert word_shape("don't") == "xxx'x"
    assert word_shape("123456789") == "dddd"        # run capped at 4
    assert word_shape("aaaaaaaa") == "xxxx"
    assert word_shape("aaaaB") == "xxxxX"           # cap resets on class change
    assert word_shape("x" * 100) == "LONG"
    assert word_shape("") == ""


def test_attr_strings_contract():
    from spacy_ray_amd.vocab.attrs import attr_strings

    norm, prefix, suffix, shape = attr_strings("Apple")
    assert norm == "apple" and prefix == "A" and suffix == "ple"
    assert shape == "Xxxxx"
    # short words: suffix is the whole word, prefix first char
    assert attr_strings("ab")[1:3] == ["a", "ab"]


def test_stringstore_roundtrip_and_select_pipes():
    from spacy_ray_amd.vocab.strings import StringStore

    ss = StringStore(["hello", "café"])
    h = ss["hello"]
    assert ss[h] == "hello"
    assert "hello" in ss and "missing" not in ss
    hs = ss.add_batch(["a", "b", "a"])
    assert ss[int(hs[0])] == "a" and len(hs) == 3

    # select_pipes freezes components for the context duration
    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.pipeline.language import init_nlp
    from tests.test_pipeline import TAGGER_CFG

    nlp = init_nlp(Config.from_str(TAGGER_CFG), sample_size=8)
    with nlp.select_pipes(disable=["tagger"]):
        assert nlp._frozen == ["tagger"]
    assert nlp._frozen == []
