#!/usr/bin/env python3
"""Flagship benchmark: en_core CNN (tok2vec+tagger+parser+NER) training
words/sec, whole node — the BASELINE.json metric.

    python bench.py --gpus N --steps K --warmup W

Driver contract: for N>1 this is launched under torch.distributed.run with
one rank per GPU (RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* in env).  W untimed
warmup steps, then exactly K timed steps bracketed by barrier +
torch.cuda.synchronize on both sides; time is MAX over ranks; rank 0 prints
one JSON line.  Synthetic data (no network for datasets), random-init
weights, bf16 on GPU.  Weak scaling: per-GPU work is fixed as N grows.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch


def build_batches(nlp, *, batch_words: int, n_batches: int, seed: int,
                  words_per_doc: int, vocab_size: int,
                  n_tags: int = 50, n_deps: int = 40, n_ent_types: int = 4):
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Example

    docs = make_synthetic_docs(
        nlp.vocab,
        n_docs=max(64, (batch_words * n_batches) // max(1, words_per_doc)),
        words_per_doc=words_per_doc,
        vocab_size=vocab_size,
        n_tags=n_tags,
        n_deps=n_deps,
        n_ent_types=n_ent_types,
        seed=seed,
        world_seed=0,
    )
    batches = []
    cur, n = [], 0
    for d in docs:
        cur.append(Example.from_doc(d))
        n += len(d)
        if n >= batch_words:
            batches.append(cur)
            cur, n = [], 0
            if len(batches) >= n_batches:
                break
    if not batches:
        batches = [[Example.from_doc(d) for d in docs]]
    return batches


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=12)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch-words", type=int, default=0,
                    help="words per step per GPU (weak scaling); default "
                         "1M on GPU, 8k on CPU")
    ap.add_argument("--words-per-doc", type=int, default=20)
    ap.add_argument("--vocab-size", type=int, default=5000)
    ap.add_argument("--config", type=str, default="examples/configs/en_core_cnn.cfg")
    ap.add_argument("--profile-steps", type=int, default=0,
                    help="if >0, run only this many unsynchronized steps (for rocprof)")
    ap.add_argument("--sweep", action="store_true",
                    help="batch-size sweep (3k/32k/128k/1M words): one JSON "
                         "line per size — the realistic-batch curve next to "
                         "the 1M-word headline (VERDICT r1 item 8)")
    args = ap.parse_args()

    if args.sweep:
        import subprocess

        for bw in (3000, 32768, 131072, 1000000):
            cmd = [sys.executable, __file__, "--steps", str(args.steps),
                   "--warmup", str(args.warmup), "--batch-words", str(bw),
                   "--config", args.config]
            subprocess.run(cmd, check=True)
        return

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import init_comm_from_env
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp

    use_cuda = torch.cuda.is_available()
    if args.batch_words == 0:
        args.batch_words = 1000000 if use_cuda else 8000
    if use_cuda:
        torch.cuda.set_device(local_rank)
        device = f"cuda:{local_rank}"
    else:
        device = "cpu"
    comm = init_comm_from_env()

    here = os.path.dirname(os.path.abspath(__file__))
    cfg_path = args.config if os.path.isabs(args.config) else os.path.join(here, args.config)
    config = Config.from_disk(cfg_path)
    nlp = init_nlp(config, device=device, sample_size=64)
    T = resolve(config.interpolate()["training"], validate=False)
    engine = ZeRO1Engine(nlp, T["optimizer"], comm)
    torch.manual_seed(1234 + rank)

    # label spaces must match the config's training corpus (the parser/NER
    # gold mapping fails loudly on labels outside the discovered set)
    label_space = {
        "xx_multilingual.cfg": dict(n_tags=17, n_deps=37, n_ent_types=1),
    }.get(os.path.basename(args.config), {})
    batches = build_batches(
        nlp, batch_words=args.batch_words,
        n_batches=8 if args.batch_words <= 64000 else 4,
        seed=100 + rank, words_per_doc=args.words_per_doc,
        vocab_size=args.vocab_size, **label_space,
    )
    words_per_step = sum(len(eg) for eg in batches[0])
    # fixed replayed batches: precompute the device-side TokenBatch once
    from spacy_ray_amd.models.batch import TokenBatch

    token_batches = [TokenBatch([eg.predicted for eg in b], device) for b in batches]

    def step(i: int) -> None:
        k = i % len(batches)
        engine.accumulate(batches[k], drop=0.1, token_batch=token_batches[k])
        engine.apply_step()

    if args.profile_steps:
        for i in range(args.profile_steps):
            step(i)
        if use_cuda:
            torch.cuda.synchronize()
        return

    # warmup must cover EVERY distinct batch shape: hipBLASLt runs solution
    # selection per new GEMM M, which otherwise lands in the timed region.
    # The count must be RANK-UNIFORM: every step issues the same fixed
    # sequence of collectives, so ranks running different warmup counts
    # (len(batches) varies with the rank-seeded doc lengths) deadlock —
    # one rank's reduce-scatter pairs with another's barrier.  Take the
    # global max.
    n_warm = max(args.warmup, len(batches))
    if world > 1:
        import torch.distributed as dist

        t = torch.tensor([float(n_warm)], dtype=torch.float64)
        t_dev = t.to(device) if comm.backend == "nccl" else t
        dist.all_reduce(t_dev, op=dist.ReduceOp.MAX)
        n_warm = int(t_dev.item())
    for i in range(n_warm):
        step(i)
    if os.environ.get("SRX_TIMING") == "1":
        from spacy_ray_amd.utils import timing

        timing.reset_times()  # drop warmup/first-call one-time costs

    comm.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    if use_cuda:
        torch.cuda.synchronize()
    comm.barrier()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    t = torch.tensor([elapsed], dtype=torch.float64)
    if world > 1:
        import torch.distributed as dist

        t_dev = t.to(device) if comm.backend == "nccl" else t
        dist.all_reduce(t_dev, op=dist.ReduceOp.MAX)
        elapsed = float(t_dev.item())

    if rank == 0 and os.environ.get("SRX_TIMING") == "1":
        from spacy_ray_amd.utils import timing

        tt = timing.phase_times()
        cc = timing.phase_counts()
        for k in sorted(tt):
            print(f"# {k}: {tt[k]:.0f} ms total, {cc[k]} calls", flush=True)

    cfg_base = os.path.basename(args.config)
    model_desc = {
        "en_core_cnn.cfg": ("words/sec (whole node) en_core CNN tagger+parser+NER train",
                            "en_core_web_cnn (MultiHashEmbed+MaxoutWindowEncoder w96d4 + tagger + parser + ner)"),
        "en_core_trf_hash.cfg": ("words/sec (whole node) en_core_web_trf (roberta-base, hash subwords) train",
                                 "en_core_web_trf shape (roberta-base random-init, 1 subword/word)"),
        "en_core_trf.cfg": ("words/sec (whole node) en_core_web_trf (roberta-base) train",
                            "en_core_web_trf (roberta-base random-init + tagger + parser + ner)"),
        "xx_multilingual.cfg": ("words/sec (whole node) xx multilingual UD train",
                                "xx multilingual UD (MultiHashEmbed w128d4 + tagger + parser, 8 treebanks)"),
    }.get(cfg_base, ("words/sec (whole node) " + cfg_base, cfg_base))

    if rank == 0:
        total_words = words_per_step * world * args.steps
        value = total_words / elapsed
        out = {
            "metric": model_desc[0],
            "value": value,
            "unit": "words/s",
            "n_gpus": world if use_cuda else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": model_desc[1],
                "global_batch": words_per_step * world,
                "seq_len": args.words_per_doc,
                "parallelism": f"dp{world}",
            },
        }
        print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
