"""HTTP serving: build_app over a trained pipeline, annotate texts through
the real decode path (TestClient, no sockets)."""
import os
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


@pytest.fixture(scope="module")
def served_nlp():
    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.pipeline.language import init_nlp

    cfg = Config.from_disk(REPO / "examples" / "configs" / "en_core_cnn.cfg")
    return init_nlp(cfg, device="cpu", sample_size=16)


def test_serve_endpoints(served_nlp):
    from fastapi.testclient import TestClient

    from spacy_ray_amd.serve.app import build_app

    client = TestClient(build_app(served_nlp))
    assert client.get("/health").json() == {"status": "ok"}

    info = client.get("/info").json()
    assert info["pipeline"] == served_nlp.pipe_names
    assert "tagger" in info["labels"]

    r = client.post("/annotate", json={"texts": ["hello brave new world", "x y"]})
    assert r.status_code == 200, r.text
    docs = r.json()["docs"]
    assert len(docs) == 2
    d0 = docs[0]
    assert d0["words"] == ["hello", "brave", "new", "world"]
    assert len(d0["tags"]) == 4
    assert len(d0["heads"]) == 4
    assert len(d0["ents"]) == 4 and all(isinstance(t, str) for t in d0["ents"])
    assert isinstance(d0["spans"], list)

    # empty batch is fine
    r = client.post("/annotate", json={"texts": []})
    assert r.status_code == 200 and r.json()["docs"] == []


def test_serve_spans_match_biluo(served_nlp):
    from fastapi.testclient import TestClient

    from spacy_ray_amd.serve.app import build_app
    from spacy_ray_amd.train.scorer import _ents_to_spans

    client = TestClient(build_app(served_nlp))
    r = client.post("/annotate", json={"texts": ["alpha beta gamma delta epsilon"]})
    d = r.json()["docs"][0]
    expect = sorted(_ents_to_spans(d["ents"]))
    got = [(s["start"], s["end"], s["label"]) for s in d["spans"]]
    assert got == expect


def test_package_load_roundtrip(tmp_path):
    """spacy_ray_amd.load() on a saved checkpoint annotates text."""
    import spacy_ray_amd
    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.pipeline.language import init_nlp

    cfg = Config.from_disk(REPO / "examples" / "configs" / "en_tagger_cpu.cfg")
    nlp = init_nlp(cfg, device="cpu", sample_size=8)
    nlp.to_disk(tmp_path / "model")
    nlp2 = spacy_ray_amd.load(tmp_path / "model")
    doc = nlp2("hello world again")
    assert doc.tags and len(doc.tags) == 3


def test_micro_batcher_coalesces_concurrent_requests():
    """Concurrent submits within the wait window share ONE model call and
    results split back per request, in order."""
    import asyncio

    from spacy_ray_amd.serve.app import MicroBatcher

    calls = []

    def run_fn(texts):
        calls.append(list(texts))
        return [t.upper() for t in texts]

    async def main():
        b = MicroBatcher(run_fn, max_batch=100, max_wait_ms=200)
        r = await asyncio.gather(
            b.submit(["a", "b"]), b.submit(["c"]), b.submit(["d", "e"]))
        return r, b

    results, b = asyncio.run(main())
    assert results == [["A", "B"], ["C"], ["D", "E"]]
    assert len(calls) == 1 and calls[0] == ["a", "b", "c", "d", "e"]
    assert b.batches_run == 1 and b.requests_served == 3


def test_micro_batcher_propagates_errors():
    import asyncio

    import pytest

    from spacy_ray_amd.serve.app import MicroBatcher

    def boom(texts):
        raise RuntimeError("model exploded")

    async def main():
        b = MicroBatcher(boom, max_wait_ms=10)
        with pytest.raises(RuntimeError, match="exploded"):
            await b.submit(["x"])
        # batcher survives: a healthy fn via a fresh batcher works after
        return True

    assert asyncio.run(main())


def test_micro_batcher_respects_max_batch():
    import asyncio

    from spacy_ray_amd.serve.app import MicroBatcher

    calls = []

    def run_fn(texts):
        calls.append(len(texts))
        return list(texts)

    async def main():
        b = MicroBatcher(run_fn, max_batch=2, max_wait_ms=200)
        return await asyncio.gather(
            b.submit(["1", "2"]), b.submit(["3", "4"]))

    out = asyncio.run(main())
    assert out == [["1", "2"], ["3", "4"]]
    assert calls == [2, 2]  # two groups, not one of 4


def test_annotate_empty_and_whitespace_texts(served_nlp):
    from fastapi.testclient import TestClient

    from spacy_ray_amd.serve.app import build_app

    client = TestClient(build_app(served_nlp))
    r = client.post("/annotate", json={"texts": ["", "  ", "ok then"]})
    assert r.status_code == 200
    docs = r.json()["docs"]
    assert docs[0]["words"] == [] and docs[1]["words"] == []
    assert docs[2]["words"] == ["ok", "then"]
