"""Per-rank training worker.

The MI355X re-architecture of the reference Worker actor
(`/root/reference/spacy_ray/worker.py:23-262`): one OS process per GPU
(launched by parallel/launcher.py or torchrun), RCCL process group instead of
Ray RPC endpoints, synchronous ZeRO-1 steps instead of the async proxy.
Responsibilities kept 1:1: build nlp from config, resolve
the training schema, run the train_while_improving iterator, rank-0 eval +
score broadcast (the Evaluator actor's role, worker.py:281-300 -> C3
broadcast), rank-0 checkpointing (wired, unlike the reference's TODO at
train_cli.py:41), cluster-scaled words logging (worker.py:308-311).
"""
from __future__ import annotations

import json
import logging
import os
import time
from pathlib import Path
from typing import Dict, Iterator, Optional

import torch

from spacy_ray_amd.config.config import Config, resolve, resolve_dot_names
from spacy_ray_amd.parallel.comm import Comm, init_comm_from_env
from spacy_ray_amd.parallel.engine import ZeRO1Engine
from spacy_ray_amd.pipeline.language import init_nlp
from spacy_ray_amd.train.loop import create_train_batches, train_while_improving
from spacy_ray_amd.train.scorer import weighted_score


def _shard_corpus(corpus, rank: int, world: int):
    """Explicit rank::world interleave so an epoch is a true partition
    (improves on the reference, where every worker iterates the full corpus —
    SURVEY.md §2.3 DP row).  Every rank yields exactly ceil(n/world)
    examples (wrapping on the remainder) so per-rank epochs stay aligned."""
    if world <= 1:
        return corpus

    def sharded(nlp) -> Iterator:
        egs = list(corpus(nlp))
        n = len(egs)
        if n == 0:
            return
        per = -(-n // world)
        for j in range(per):
            yield egs[(j * world + rank) % n]

    return sharded


def _synced_batches(batches, comm: Comm, device) -> Iterator:
    """Stop ALL ranks as soon as ANY rank's batch stream ends.  Word-count
    batching can split a sharded epoch into different batch counts per rank;
    without this agreement one rank exits the loop while the others wait in
    the eval broadcast (finite-epoch hang).  One tiny all-reduce per step."""
    if comm.world <= 1:
        yield from batches
        return
    it = iter(batches)
    dev = device if str(device).startswith("cuda") else "cpu"
    while True:
        try:
            item = next(it)
            has = 1.0
        except StopIteration:
            item = None
            has = 0.0
        flag = torch.tensor([has], device=dev)
        comm.all_reduce_(flag)
        if float(flag.item()) < comm.world:
            return
        yield item


def _check_param_manifest(nlp, comm: Comm) -> None:
    """Assert identical parameter (name, shape) tables across ranks — the
    cross-rank key invariant."""
    manifest = [
        (n, tuple(p.shape))
        for n, p in nlp.torch_module().named_parameters()
    ]
    ref = comm.broadcast_obj(manifest, src=0)
    if manifest != ref:
        raise RuntimeError(
            f"rank {comm.rank}: parameter manifest differs from rank 0 — "
            "non-deterministic model build"
        )


def distributed_train(
    config: Config,
    *,
    output_path: Optional[Path] = None,
    use_gpu: int = -1,
    code_path: Optional[Path] = None,
    resume: bool = False,
    metrics_path: Optional[Path] = None,
) -> Dict:
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    if code_path:
        import importlib.util

        spec = importlib.util.spec_from_file_location("srx_user_code", code_path)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)

    if use_gpu >= 0 and torch.cuda.is_available():
        n_dev = torch.cuda.device_count()
        if local_rank >= n_dev:
            raise SystemExit(
                f"rank {rank}: LOCAL_RANK {local_rank} needs GPU {local_rank} "
                f"but only {n_dev} visible — --n-workers must be <= GPUs per "
                f"node (or restrict with HIP_VISIBLE_DEVICES)"
            )
        device = f"cuda:{local_rank}"
        torch.cuda.set_device(local_rank)
    else:
        device = "cpu"

    comm = init_comm_from_env()
    icfg = config.interpolate()
    from spacy_ray_amd.config.schemas import ConfigSchemaTraining

    T = resolve(icfg["training"], schema=ConfigSchemaTraining)

    nlp = init_nlp(config, device=device)
    init_t2v = icfg.get("training", {}).get("init_tok2vec")
    if init_t2v and not resume:
        # spaCy's [initialize] init_tok2vec contract: load pretrained
        # encoder weights (spacy-mi pretrain output) before the engine
        # snapshots the fp32 master.  Same file on every rank => params
        # stay rank-identical.
        from spacy_ray_amd.train.pretrain import load_init_tok2vec

        n_loaded = load_init_tok2vec(nlp, init_t2v)
        if rank == 0:
            logging.getLogger("spacy_ray_amd").info(
                "loaded %d pretrained tok2vec tensors from %s",
                n_loaded, init_t2v)
    _check_param_manifest(nlp, comm)
    # params are now identical on all ranks; diverge the RNG for dropout
    seed = int(icfg.get("training", {}).get("seed", 0) or 0)
    torch.manual_seed(seed * 1000 + 17 * rank + 1)

    train_corpus, dev_corpus = resolve_dot_names(
        icfg, [T["train_corpus"], T["dev_corpus"]]
    )
    train_corpus = _shard_corpus(train_corpus, rank, world)
    # frozen components: exclude their params from the flat buffer BEFORE the
    # engine flattens (spaCy contract: frozen pipes are not updated at all —
    # leaving them in would apply AdamW's decoupled weight decay every step
    # even with zero gradient, silently shrinking the frozen weights)
    for pname in T.get("frozen_components") or []:
        pipe = dict(nlp.pipeline).get(pname)
        if pipe is not None and getattr(pipe, "module", None) is not None:
            for p in pipe.module.parameters():
                p.requires_grad_(False)
    engine = ZeRO1Engine(nlp, T["optimizer"], comm)
    if resume and output_path and (Path(output_path) / "model-last").exists():
        nlp.from_disk(Path(output_path) / "model-last")
        opt_state = Path(output_path) / "model-last" / f"optim.rank{rank}.pt"
        if opt_state.exists():
            engine.load_state_dict(torch.load(opt_state, map_location=device))
        else:
            # params loaded but no optimizer shard: re-snapshot the master
            # so the first step doesn't revert to pre-load weights
            engine.refresh_master_from_params()

    # before_to_disk: optional @callbacks hook applied to nlp before saving
    # (contract of create_before_to_disk_callback at
    # /root/reference/spacy_ray/worker.py:96,222)
    before_to_disk = T.get("before_to_disk")

    # resume: previous best composite score (written into model-best's
    # meta.json) seeds the loop so a worse resumed model never
    # overwrites model-best
    prev_best = None
    if resume and output_path:
        best_meta = Path(output_path) / "model-best" / "meta.json"
        if best_meta.exists():
            prev_best = json.loads(best_meta.read_text()).get("best_score")

    dev_examples = None
    eval_state: Dict[str, bool] = {}

    def evaluate():
        """Sharded dev-set evaluation: every rank decodes its rank::world
        slice and the additive score counts are all-gathered + merged
        (VERDICT r1 item 9 — rank-0-serial eval idled N-1 GPUs at every
        checkpoint).  use_averages: the running parameter average is
        swapped in on ALL ranks (the swap all-gathers)."""
        nonlocal dev_examples
        from spacy_ray_amd.train.scorer import counts_to_scores, merge_counts, score_counts

        with engine.averaged_params():
            if dev_examples is None:
                dev_examples = list(dev_corpus(nlp))[rank::world] if world > 1 \
                    else list(dev_corpus(nlp))
            t_eval = time.time()
            batch_size = 256
            for i in range(0, len(dev_examples), batch_size):
                chunk = dev_examples[i : i + batch_size]
                nlp.predict_docs([eg.predicted for eg in chunk])
            counts = score_counts(dev_examples, nlp.pipe_names)
            counts["_words"] = sum(len(eg) for eg in dev_examples)
            dt = max(1e-9, time.time() - t_eval)
        parts = comm.all_gather_obj((counts, dt)) if world > 1 else [(counts, dt)]
        merged = merge_counts([p[0] for p in parts])
        scores = counts_to_scores(merged)
        scores["speed"] = merged.get("_words", 0) / max(p[1] for p in parts)
        weights = T.get("score_weights") or {}
        if rank == 0:
            missing = [k for k, w in weights.items() if w and k not in scores]
            if missing and not eval_state.get("warned_weights"):
                # the reference errors here (E983, loggers.py:30-37); we
                # warn once so a typo'd weight key can't silently zero
                # the model-best selection
                logging.getLogger(__name__).warning(
                    "score_weights keys %s not produced by evaluate "
                    "(have: %s) — they contribute 0 to the composite",
                    missing, sorted(scores))
                eval_state["warned_weights"] = True
        return weighted_score(scores, weights), scores

    batches = _synced_batches(
        create_train_batches(nlp, train_corpus, T["batcher"], T.get("max_epochs", 0) or 0),
        comm, device,
    )
    if rank == 0:
        logger_setup = T.get("logger")
        if logger_setup is None:
            from spacy_ray_amd.train.loggers import console_logger

            logger_setup = console_logger()
        print_row, finalize_logger = logger_setup(nlp)
    else:
        print_row, finalize_logger = (lambda info: None), (lambda: None)

    metrics_fh = open(metrics_path, "a") if (metrics_path and rank == 0) else None
    words_cum = 0
    t_start = time.time()
    t_last = t_start
    final_info: Dict = {}

    step_iter = train_while_improving(
        nlp,
        engine,
        batches,
        evaluate=evaluate,
        dropout=T.get("dropout", 0.1),
        accumulate_gradient=int(T.get("accumulate_gradient", 1) or 1),
        patience=int(T.get("patience", 0) or 0),
        max_steps=int(T.get("max_steps", 0) or 0),
        eval_frequency=int(T.get("eval_frequency", 200) or 200),
        exclude=T.get("frozen_components") or [],
        annotating_components=T.get("annotating_components") or [],
        before_update=T.get("before_update"),
        initial_best=prev_best,
    )
    # test hook: SRX_FAULT_INJECT="rank:step" kills this
    # rank at that step so the supervisor's all-ranks-abort path is testable
    fault = os.environ.get("SRX_FAULT_INJECT")
    fault_rank, fault_step = (int(x) for x in fault.split(":")) if fault else (-1, -1)

    for batch, info, is_best_checkpoint in step_iter:
        if rank == fault_rank and info["step"] == fault_step:
            raise SystemExit(41)
        words_cum += info["words"] * world
        final_info = info
        if rank == 0:
            now = time.time()
            info["words_scaled"] = info["words"] * world
            info["wps"] = info["words"] * world / max(1e-9, now - t_last)
            t_last = now
            if metrics_fh is not None:
                metrics_fh.write(json.dumps({
                    "step": info["step"], "epoch": info["epoch"],
                    "losses": info["losses"], "score": info["score"],
                    "other_scores": info["other_scores"],
                    "words": info["words_scaled"], "words_cum": words_cum,
                    "wps": info["wps"], "time": now - t_start,
                    "compute_ms": round(getattr(engine, "last_compute_ms", 0.0), 2),
                    "comm_opt_ms": round(getattr(engine, "last_comm_ms", 0.0), 2),
                }) + "\n")
                metrics_fh.flush()
            if is_best_checkpoint is not None:
                print_row(info)
        if is_best_checkpoint and output_path:
            # all ranks participate: params are replicated post-all-gather;
            # rank 0 writes the pipeline, every rank its optimizer shard.
            # Saved under the AVERAGED params when use_averages is on (the
            # spaCy `with nlp.use_params(optimizer.averages)` contract) so
            # the checkpoint reproduces the eval score; no-op otherwise.
            with engine.averaged_params():
                if rank == 0:
                    nlp.meta["performance"] = info["other_scores"]
                    nlp.meta["best_score"] = info["score"]
                    to_save = before_to_disk(nlp) if before_to_disk else nlp
                    to_save.to_disk(Path(output_path) / "model-best")
                comm.barrier()
    if output_path:
        with engine.averaged_params():
            if rank == 0:
                to_save = before_to_disk(nlp) if before_to_disk else nlp
                to_save.to_disk(Path(output_path) / "model-last")
            comm.barrier()
        torch.save(engine.state_dict(), Path(output_path) / "model-last" / f"optim.rank{rank}.pt")
    if rank == 0:
        finalize_logger()
        if metrics_fh is not None:
            metrics_fh.close()
    return final_info
