"""HTTP serving for trained pipelines (FastAPI).

The reference is training-only (SURVEY.md §2 has no serving surface); this
module is framework surface for the production-deployment goal: load a
spaCy-layout checkpoint, annotate batches of texts through the GPU decode
path (`Language.pipe` -> interleaved transition decode), return JSON docs.

Endpoints:
  GET  /health       liveness
  GET  /info         pipeline names, labels per pipe, device
  POST /annotate     {"texts": [...]} -> {"docs": [{words, tags, heads,
                     deps, ents, spans}, ...]} — `spans` is the
                     entity-span view ([start, end, label] in tokens)
                     derived from the per-token BILUO tags.

Concurrency: one model instance guarded by a lock (FastAPI runs sync
endpoints on a threadpool; the decode path mutates per-call state machines
but shares module weights).  Scale-out story is one uvicorn process per
GPU behind a load balancer — same one-process-per-GPU shape as training.
"""
import threading
from typing import List

from pydantic import BaseModel

from spacy_ray_amd.train.scorer import _ents_to_spans


class AnnotateRequest(BaseModel):
    # module level (not inside build_app): FastAPI resolves endpoint
    # annotations via get_type_hints against module globals
    texts: List[str]


def build_app(nlp, max_batch: int = 256):
    from fastapi import FastAPI

    app = FastAPI(title="spacy-mi", version="0.1.0")
    lock = threading.Lock()

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
        }

    @app.post("/annotate")
    def annotate(req: AnnotateRequest):
        with lock:
            docs = list(nlp.pipe(req.texts, batch_size=max_batch))
        out = []
        for d in docs:
            rec = d.to_dict()
            rec["spans"] = [
                {"start": s, "end": e, "label": lab}
                for (s, e, lab) in sorted(_ents_to_spans(d.ents or []))
            ]
            out.append(rec)
        return {"docs": out}

    return app


def serve(model_path, host: str = "127.0.0.1", port: int = 8000,
          use_gpu: int = -1, max_batch: int = 256) -> None:
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
    uvicorn.run(build_app(nlp, max_batch=max_batch), host=host, port=port)
