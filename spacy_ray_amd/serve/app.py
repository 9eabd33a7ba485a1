"""HTTP serving for trained pipelines (FastAPI).

The reference is training-only (SURVEY.md §2 has no serving surface); this
module is framework surface for the production-deployment goal: load a
spaCy-layout checkpoint, annotate batches of texts through the GPU decode
path (`Language.pipe` -> interleaved transition decode), return JSON docs.

Endpoints:
  GET  /health       liveness
  GET  /info         pipeline names, labels per pipe, device, batch stats
  POST /annotate     {"texts": [...]} -> {"docs": [{words, tags, heads,
                     deps, ents, spans}, ...]} — `spans` is the
                     entity-span view ([start, end, label] in tokens)
                     derived from the per-token BILUO tags.

Concurrency: requests are coalesced by a dynamic micro-batcher — an
asyncio queue drains concurrent requests into ONE decode batch (up to
`max_batch` docs or `max_wait_ms`), so many small requests share a single
GPU pass instead of serializing N tiny ones.  The model itself runs on a
single executor thread (one model instance per process).  Scale-out story
is one uvicorn process per GPU behind a load balancer — the same
one-process-per-GPU shape as training.
"""
import asyncio
import threading
from typing import List

from pydantic import BaseModel

from spacy_ray_amd.train.scorer import _ents_to_spans


class AnnotateRequest(BaseModel):
    # module level (not inside build_app): FastAPI resolves endpoint
    # annotations via get_type_hints against module globals
    texts: List[str]


class MicroBatcher:
    """Coalesce concurrent annotate requests into shared decode batches.

    First request of a group is served immediately-ish: the collector waits
    at most `max_wait_ms` for followers, caps the group at `max_batch`
    docs, runs `run_fn` once on the concatenation, then splits the results
    back per request.  Exceptions propagate to every request in the group.
    """

    def __init__(self, run_fn, max_batch: int = 256, max_wait_ms: float = 5.0):
        self.run_fn = run_fn
        self.max_batch = max_batch
        self.max_wait = max_wait_ms / 1000.0
        self.queue: "asyncio.Queue" = asyncio.Queue()
        self.batches_run = 0
        self.requests_served = 0
        self._collector = None
        self._lock = threading.Lock()  # the model is single-instance

    async def submit(self, texts: List[str]) -> list:
        if self._collector is None or self._collector.done():
            self._collector = asyncio.get_running_loop().create_task(self._collect())
        fut = asyncio.get_running_loop().create_future()
        await self.queue.put((texts, fut))
        return await fut

    async def _collect(self) -> None:
        loop = asyncio.get_running_loop()
        while True:
            texts, fut = await self.queue.get()
            group = [(texts, fut)]
            n = len(texts)
            deadline = loop.time() + self.max_wait
            while n < self.max_batch:
                timeout = deadline - loop.time()
                if timeout <= 0:
                    break
                try:
                    item = await asyncio.wait_for(self.queue.get(), timeout)
                except asyncio.TimeoutError:
                    break
                group.append(item)
                n += len(item[0])
            flat = [t for texts_, _ in group for t in texts_]

            def run():
                with self._lock:
                    return self.run_fn(flat)

            try:
                docs = await loop.run_in_executor(None, run)
            except Exception as e:  # propagate to every waiter in the group
                for _, f in group:
                    if not f.done():
                        f.set_exception(e)
                continue
            self.batches_run += 1
            self.requests_served += len(group)
            off = 0
            for texts_, f in group:
                if not f.done():
                    f.set_result(docs[off:off + len(texts_)])
                off += len(texts_)


def _doc_json(d) -> dict:
    rec = d.to_dict()
    # Doc.spans groups (spancat output) under their own key; "spans" stays
    # the entity-span view derived from BILUO tags (response compat)
    rec["span_groups"] = rec.pop("spans", None)
    rec["spans"] = [
        {"start": s, "end": e, "label": lab}
        for (s, e, lab) in sorted(_ents_to_spans(d.ents or []))
    ]
    return rec


def build_app(nlp, max_batch: int = 256, max_wait_ms: float = 5.0):
    from fastapi import FastAPI

    app = FastAPI(title="spacy-mi", version="0.1.0")
    batcher = MicroBatcher(
        lambda texts: list(nlp.pipe(texts, batch_size=max_batch)),
        max_batch=max_batch, max_wait_ms=max_wait_ms,
    )
    app.state.batcher = batcher

    @app.get("/health")
    def health():
        return {"status": "ok"}

    @app.get("/info")
    def info():
        return {
            "pipeline": nlp.pipe_names,
            "device": str(nlp.device),
            "labels": {
                name: list(getattr(pipe, "labels", []) or [])
                for name, pipe in nlp.pipeline
            },
            "batches_run": batcher.batches_run,
            "requests_served": batcher.requests_served,
        }

    @app.post("/annotate")
    async def annotate(req: AnnotateRequest):
        if not req.texts:
            return {"docs": []}
        docs = await batcher.submit(req.texts)
        return {"docs": [_doc_json(d) for d in docs]}

    return app


def serve(model_path, host: str = "127.0.0.1", port: int = 8000,
          use_gpu: int = -1, max_batch: int = 256,
          max_wait_ms: float = 5.0) -> None:
    """Load a checkpoint directory and serve it (blocking)."""
    import torch
    import uvicorn

    import spacy_ray_amd

    device = (
        f"cuda:{max(use_gpu, 0)}"
        if (use_gpu >= 0 and torch.cuda.is_available())
        else "cpu"
    )
    nlp = spacy_ray_amd.load(model_path, device=device)
    uvicorn.run(build_app(nlp, max_batch=max_batch, max_wait_ms=max_wait_ms),
                host=host, port=port)
