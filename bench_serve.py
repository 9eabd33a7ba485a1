"""Decode-throughput benchmark: words/sec through Language.pipe (the serve
path) on synthetic docs — the inference-side counterpart of bench.py.

Not part of the driver's bench contract (bench.py measures training); run
manually:

    python bench_serve.py [--config examples/configs/en_core_cnn.cfg]
                          [--batch-words 200000] [--steps 10] [--warmup 3]
                          [--gpu-id 0]

Prints one JSON line: decode words/sec over the full pipeline
(tok2vec forward + tagger argmax + parser/NER greedy decode).
"""
from __future__ import annotations

import argparse
import json
import os
import time


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="examples/configs/en_core_cnn.cfg")
    ap.add_argument("--batch-words", type=int, default=0,
                    help="words per decode batch (default: 200k GPU / 4k CPU)")
    ap.add_argument("--words-per-doc", type=int, default=20)
    ap.add_argument("--vocab-size", type=int, default=5000)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--gpu-id", type=int, default=-1)
    args = ap.parse_args()

    import torch

    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.pipeline.language import init_nlp

    use_cuda = torch.cuda.is_available() and (args.gpu_id >= 0 or os.environ.get("SRX_BENCH_GPU") == "1")
    if torch.cuda.is_available() and args.gpu_id < 0:
        use_cuda = True  # a GPU box means a GPU bench unless told otherwise
    device = f"cuda:{max(args.gpu_id, 0)}" if use_cuda else "cpu"
    batch_words = args.batch_words or (200_000 if use_cuda else 4_000)

    here = os.path.dirname(os.path.abspath(__file__))
    cfg_path = args.config if os.path.isabs(args.config) else os.path.join(here, args.config)
    nlp = init_nlp(Config.from_disk(cfg_path), device=device, sample_size=64)

    n_docs = max(1, batch_words // args.words_per_doc)
    docs = make_synthetic_docs(nlp.vocab, n_docs=n_docs,
                               words_per_doc=args.words_per_doc,
                               vocab_size=args.vocab_size, n_tags=50,
                               n_deps=40, n_ent_types=4, seed=7)
    total_words = sum(len(d) for d in docs)

    def run_once():
        nlp.predict_docs([d.copy_unannotated() for d in docs])

    for _ in range(args.warmup):
        run_once()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_once()
    if use_cuda:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    print(json.dumps({
        "metric": "decode words/sec (predict_docs full pipeline)",
        "value": total_words * args.steps / dt,
        "unit": "words/s",
        "ms_per_batch": dt / args.steps * 1000,
        "batch_words": total_words,
        "steps": args.steps,
        "warmup": args.warmup,
        "device": device,
        "config": os.path.basename(cfg_path),
        "data": "synthetic",
    }))


if __name__ == "__main__":
    main()
