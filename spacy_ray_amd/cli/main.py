"""CLI: ``spacy-mi ray train config.cfg --n-workers N [...]``.

Flag surface of the reference command (`/root/reference/spacy_ray/
train_cli.py:26-53`): config_path positional; --code, --output, --n-workers,
--address, --gpu-id, --verbose; trailing ``--dotted.key value`` config
overrides.  ``--output`` is actually wired here (the reference parses it and
drops it, train_cli.py:41).  The command is the launcher (L5): it spawns one
worker process per GPU/rank and supervises them.
"""
from __future__ import annotations

import json
import logging
import os
import sys
from pathlib import Path
from typing import Optional

import typer

from spacy_ray_amd.config.config import Config, parse_config_overrides

app = typer.Typer(name="spacy-mi", no_args_is_help=True)
ray_app = typer.Typer(name="ray", no_args_is_help=True,
                      help="Distributed/parallel training (reference-compatible command group)")
app.add_typer(ray_app)


@ray_app.command(
    "train",
    context_settings={"allow_extra_args": True, "ignore_unknown_options": True},
)
def ray_train_cli(
    ctx: typer.Context,
    config_path: Path = typer.Argument(..., help="Path to config file"),
    code_path: Optional[Path] = typer.Option(None, "--code", "-c", help="Path to Python file with additional code to be imported"),
    output_path: Optional[Path] = typer.Option(None, "--output", "-o", help="Output directory for checkpoints"),
    n_workers: int = typer.Option(1, "--n-workers", "-n", help="Number of workers (1 process per GPU)"),
    address: Optional[str] = typer.Option(None, "--address", "-a", help="Rendezvous address host[:port] (multi-node)"),
    use_gpu: int = typer.Option(-1, "--gpu-id", "-g", help="GPU ID or -1 for CPU"),
    verbose: bool = typer.Option(False, "--verbose", "-V", help="Display more information"),
    resume: bool = typer.Option(False, "--resume", help="Resume from <output>/model-last (params + per-rank optimizer shards)"),
    init_tok2vec: Optional[Path] = typer.Option(None, "--init-tok2vec", help="Pretrained tok2vec weights (spacy-mi pretrain output dir or .safetensors)"),
    nnodes: int = typer.Option(1, "--nnodes", help="Number of nodes (with --address on every node)"),
    node_rank: int = typer.Option(0, "--node-rank", help="This node's rank in [0, nnodes)"),
):
    """Train a pipeline with N data-parallel workers over RCCL/xGMI."""
    logging.basicConfig(level=logging.DEBUG if verbose else logging.ERROR)
    overrides = parse_config_overrides(list(ctx.args))
    if init_tok2vec is not None:
        overrides["training.init_tok2vec"] = str(init_tok2vec)
    config = Config.from_disk(config_path, overrides=overrides)
    raise SystemExit(
        ray_train(config, config_path=config_path, output_path=output_path,
                  code_path=code_path, n_workers=n_workers, address=address,
                  use_gpu=use_gpu, overrides=overrides, resume=resume,
                  nnodes=nnodes, node_rank=node_rank)
    )


def ray_train(
    config: Config,
    *,
    config_path: Path,
    output_path: Optional[Path] = None,
    code_path: Optional[Path] = None,
    n_workers: int = 1,
    address: Optional[str] = None,
    use_gpu: int = -1,
    overrides: Optional[dict] = None,
    resume: bool = False,
    nnodes: int = 1,
    node_rank: int = 0,
) -> int:
    """Launcher (contract of `/root/reference/spacy_ray/train_cli.py:56-91`)."""
    from spacy_ray_amd.parallel.launcher import launch_workers

    if output_path:
        Path(output_path).mkdir(parents=True, exist_ok=True)
    if n_workers <= 1 and not address and nnodes <= 1:
        # single process: run in-process, no process group
        from spacy_ray_amd.train.worker import distributed_train

        distributed_train(
            config,
            output_path=output_path,
            use_gpu=use_gpu,
            code_path=code_path,
            resume=resume,
            metrics_path=(Path(output_path) / "metrics.jsonl") if output_path else None,
        )
        return 0
    worker_cmd = [sys.executable, "-m", "spacy_ray_amd.cli.worker", str(config_path)]
    if output_path:
        worker_cmd += ["--output", str(output_path)]
    if code_path:
        worker_cmd += ["--code", str(code_path)]
    worker_cmd += ["--gpu-id", str(use_gpu)]
    if resume:
        worker_cmd += ["--resume"]
    if overrides:
        worker_cmd += ["--overrides-json", json.dumps(overrides)]
    master_addr, master_port = "127.0.0.1", None
    if address:
        host, _, port = address.partition(":")
        master_addr = host or "127.0.0.1"
        master_port = int(port) if port else None
    if nnodes > 1 and master_port is None:
        raise SystemExit("--nnodes > 1 requires --address host:port (a fixed rendezvous port)")
    return launch_workers(worker_cmd, n_workers, master_addr=master_addr,
                          master_port=master_port, nnodes=nnodes,
                          node_rank=node_rank)


@ray_app.command("evaluate")
def ray_evaluate_cli(
    model_path: Path = typer.Argument(..., help="Trained pipeline directory (model-best/model-last)"),
    use_gpu: int = typer.Option(-1, "--gpu-id", "-g", help="GPU ID or -1 for CPU"),
    corpus: Optional[str] = typer.Option(None, "--corpus", help="Dotted corpus name in the model's config (default: the training dev corpus)"),
    output: Optional[Path] = typer.Option(None, "--output", "-o", help="Also write the scores JSON to this file"),
):
    """Evaluate a saved pipeline on its dev corpus and print scores JSON."""
    import torch

    from spacy_ray_amd.config.config import resolve, resolve_dot_names
    from spacy_ray_amd.pipeline.language import build_nlp
    from spacy_ray_amd.train.scorer import weighted_score

    device = f"cuda:{max(use_gpu, 0)}" if (use_gpu >= 0 and torch.cuda.is_available()) else "cpu"
    config = Config.from_disk(model_path / "config.cfg")
    nlp = build_nlp(config, device=device)
    nlp.from_disk(model_path)
    icfg = config.interpolate()
    T = resolve(icfg["training"])
    dot = corpus or T.get("dev_corpus", "corpora.dev")
    (dev_corpus,) = resolve_dot_names(icfg, [dot])
    examples = list(dev_corpus(nlp))
    scores = nlp.evaluate(examples)
    scores["score"] = weighted_score(scores, T.get("score_weights") or {})
    text = json.dumps(scores, indent=2)
    if output is not None:
        output.write_text(text)
    print(text)


init_app = typer.Typer(name="init", no_args_is_help=True,
                       help="Generate starter configs")
app.add_typer(init_app)


@init_app.command("fill-config")
def init_fill_config_cli(
    base_path: Path = typer.Argument(..., help="Partial config to fill"),
    output_path: Path = typer.Argument("-", help="Output path ('-' = stdout)"),
):
    """Fill a partial config with schema defaults (the `spacy init
    fill-config` role): every [training] key the schema knows gets its
    default materialized; existing values are kept."""
    import json as _json

    from spacy_ray_amd.config.schemas import ConfigSchemaTraining

    config = Config.from_disk(base_path)
    training = dict(config.get("training", {}))
    defaults = ConfigSchemaTraining().model_dump()
    filled = 0
    for key, val in defaults.items():
        if key not in training and val is not None and key != "score_weights":
            training[key] = val
            filled += 1
    config["training"] = training
    text = config.to_str()
    if str(output_path) == "-":
        print(text)
    else:
        output_path.write_text(text)
        print(f"[+] wrote {output_path} ({filled} defaults filled)")


@init_app.command("config")
def init_config_cli(
    output_path: Path = typer.Argument(..., help="Where to write the config (use - for stdout)"),
    lang: str = typer.Option("en", "--lang", "-l"),
    pipeline: str = typer.Option("tagger,parser,ner", "--pipeline", "-p",
                                 help="Comma-separated: tagger, parser, ner"),
    arch: str = typer.Option("cnn", "--arch", help="tok2vec architecture: cnn | trf"),
    width: int = typer.Option(0, "--width", help="tok2vec width (default: 96 cnn / 768 trf)"),
    gpu: bool = typer.Option(False, "--gpu", help="Tune defaults for GPU training"),
):
    """Generate a ready-to-train config (the `spacy init config` role)."""
    from spacy_ray_amd.cli.templates import render_config

    pipes = [p.strip() for p in pipeline.split(",") if p.strip()]
    bad = [p for p in pipes if p not in
           ("tagger", "parser", "ner", "textcat",
            "textcat_multilabel", "senter", "morphologizer",
            "spancat", "entity_ruler", "attribute_ruler", "lemmatizer")]
    if bad:
        raise SystemExit(f"unknown pipeline components: {bad}")
    text = render_config(lang=lang, pipes=pipes, arch=arch, width=width, gpu=gpu)
    if str(output_path) == "-":
        print(text)
    else:
        output_path.write_text(text)
        print(f"[+] wrote {output_path} — train with: "
              f"spacy-mi ray train {output_path} --output ./model")


debug_app = typer.Typer(name="debug", no_args_is_help=True,
                        help="Validate configs and data before training")
app.add_typer(debug_app)


@debug_app.command(
    "config",
    context_settings={"allow_extra_args": True, "ignore_unknown_options": True},
)
def debug_config_cli(
    ctx: typer.Context,
    config_path: Path = typer.Argument(..., help="Path to config file"),
    code_path: Optional[Path] = typer.Option(None, "--code", "-c"),
):
    """Parse, interpolate and resolve a config; build the pipeline skeleton.
    Exits nonzero with the offending section on any error."""
    from spacy_ray_amd.config.config import resolve
    from spacy_ray_amd.config.schemas import ConfigSchemaTraining
    from spacy_ray_amd.pipeline.language import build_nlp

    if code_path:
        import importlib.util

        spec = importlib.util.spec_from_file_location("srx_user_code", code_path)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
    overrides = parse_config_overrides(list(ctx.args))
    config = Config.from_disk(config_path, overrides=overrides)
    icfg = config.interpolate()
    T = resolve(icfg.get("training", {}), schema=ConfigSchemaTraining)
    nlp = build_nlp(config)
    print(f"[+] config OK: pipeline {nlp.pipe_names}")
    print(f"[+] training resolved: optimizer={type(T['optimizer']).__name__}, "
          f"max_steps={T.get('max_steps')}, eval_frequency={T.get('eval_frequency')}, "
          f"score_weights={T.get('score_weights')}")


@debug_app.command(
    "data",
    context_settings={"allow_extra_args": True, "ignore_unknown_options": True},
)
def debug_data_cli(
    ctx: typer.Context,
    config_path: Path = typer.Argument(..., help="Path to config file"),
    code_path: Optional[Path] = typer.Option(None, "--code", "-c"),
    limit: int = typer.Option(1000, "--limit", help="Max docs to scan per corpus"),
):
    """Load the train/dev corpora and report doc/token counts and label
    inventories; warns on empty corpora or train/dev label mismatches."""
    from spacy_ray_amd.config.config import resolve, resolve_dot_names
    from spacy_ray_amd.pipeline.language import build_nlp

    if code_path:
        import importlib.util

        spec = importlib.util.spec_from_file_location("srx_user_code", code_path)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
    overrides = parse_config_overrides(list(ctx.args))
    config = Config.from_disk(config_path, overrides=overrides)
    icfg = config.interpolate()
    T = resolve(icfg["training"], validate=False)
    nlp = build_nlp(config)
    warned = False
    stats = {}
    for split, dot in (("train", T.get("train_corpus", "corpora.train")),
                       ("dev", T.get("dev_corpus", "corpora.dev"))):
        (corpus,) = resolve_dot_names(icfg, [dot])
        n_docs = n_tokens = 0
        tags, deps, ents = set(), set(), set()
        cats, morphs, span_labels = set(), set(), set()
        n_sents = n_spans = 0
        for eg in corpus(nlp):
            ref = eg.reference
            n_docs += 1
            n_tokens += len(ref)
            if ref.tags:
                tags.update(ref.tags)
            if ref.deps:
                deps.update(ref.deps)
            if ref.ents:
                ents.update(t.partition("-")[2] for t in ref.ents if t not in ("O", ""))
            if ref.cats:
                cats.update(ref.cats.keys())
            if ref.morphs:
                morphs.update(m for m in ref.morphs if m)
            if ref.sent_starts is not None:
                n_sents += int((ref.sent_starts > 0).sum())
            for group in ref.spans.values():
                n_spans += len(group)
                span_labels.update(lab for (_s, _e, lab) in group)
            if n_docs >= limit:
                break
        stats[split] = (tags, deps, ents, cats, morphs, span_labels)
        extras = []
        if cats:
            extras.append(f"{len(cats)} cats")
        if morphs:
            extras.append(f"{len(morphs)} morph classes")
        if n_sents:
            extras.append(f"{n_sents} sentence starts")
        if n_spans:
            extras.append(f"{n_spans} spans ({len(span_labels)} labels)")
        extra = (", " + ", ".join(extras)) if extras else ""
        print(f"[+] {split}: {n_docs} docs, {n_tokens} tokens, "
              f"{len(tags)} tags, {len(deps)} dep labels, "
              f"{len(ents)} entity types{extra}")
        if n_docs == 0:
            print(f"[!] {split} corpus is EMPTY")
            warned = True
    for i, kind in enumerate(("tags", "dep labels", "entity types", "cats",
                              "morph classes", "span labels")):
        only_dev = stats["dev"][i] - stats["train"][i]
        if only_dev:
            print(f"[!] {kind} in dev but never in train: {sorted(only_dev)[:10]}")
            warned = True
    raise SystemExit(1 if warned else 0)


@app.command("convert")
def convert_cli(
    input_path: Path = typer.Argument(..., help="CoNLL-U (.conllu) or IOB (.iob) file"),
    output_path: Path = typer.Argument(..., help="Output DocBin (.spacy) path"),
    fmt: Optional[str] = typer.Option(None, "--format", "-f", help="conllu | iob (default: infer from extension)"),
    tag_col: str = typer.Option("upos", "--tag-col", help="CoNLL-U tag column: upos | xpos"),
):
    """Convert a CoNLL-U / IOB corpus to a DocBin for training."""
    from spacy_ray_amd.data.convert import convert_file

    n = convert_file(input_path, output_path, fmt=fmt, tag_col=tag_col)
    print(f"wrote {n} docs -> {output_path}")


@app.command("assemble")
def assemble_cli(
    ctx: typer.Context,
    config_path: Path = typer.Argument(..., help="Path to config file"),
    output_path: Path = typer.Argument(..., help="Output pipeline directory"),
):
    """Build and initialize a pipeline from a config WITHOUT training and
    save it (spaCy's `assemble` role — e.g. rule-only pipelines, or a
    random-init starting point)."""
    from spacy_ray_amd.pipeline.language import init_nlp

    overrides = parse_config_overrides(list(ctx.args))
    config = Config.from_disk(config_path, overrides=overrides)
    nlp = init_nlp(config, device="cpu")
    nlp.to_disk(output_path)
    print(f"[+] assembled pipeline {nlp.pipe_names} -> {output_path}")


@app.command("apply")
def apply_cli(
    model_path: Path = typer.Argument(..., help="Trained pipeline directory"),
    input_path: Path = typer.Argument(..., help="Input DocBin (.spacy) or text file (one doc per line)"),
    output_path: Path = typer.Argument(..., help="Output DocBin (.spacy) with predictions"),
    use_gpu: int = typer.Option(-1, "--gpu-id", "-g", help="GPU ID or -1 for CPU"),
    batch_size: int = typer.Option(256, "--batch-size", "-b"),
):
    """Annotate a corpus with a trained pipeline and write the predictions
    as a DocBin (spaCy's `apply` role)."""
    import spacy_ray_amd
    from spacy_ray_amd.data.docbin import DocBin

    device = f"cuda:{use_gpu}" if use_gpu >= 0 else "cpu"
    nlp = spacy_ray_amd.load(str(model_path), device=device)
    if str(input_path).endswith(".spacy"):
        docs_in = list(DocBin.from_disk(input_path, nlp.vocab).get_docs(nlp.vocab))
        docs = [d.copy_unannotated() for d in docs_in]
    else:
        lines = [l for l in Path(input_path).read_text().splitlines() if l.strip()]
        docs = [nlp.tokenizer(nlp.vocab, l) for l in lines]
    out = []
    for i in range(0, len(docs), batch_size):
        out.extend(nlp.predict_docs(docs[i:i + batch_size]))
    DocBin(out).to_disk(output_path)
    print(f"[+] annotated {len(out)} docs -> {output_path}")


@debug_app.command("model")
def debug_model_cli(
    ctx: typer.Context,
    config_path: Path = typer.Argument(..., help="Path to config file"),
    component: str = typer.Argument("", help="Component name (default: all)"),
):
    """Build the pipeline skeleton and print each component's parameter
    table (name, shape, count) — spaCy's `debug model` role."""
    from spacy_ray_amd.pipeline.language import init_nlp

    overrides = parse_config_overrides(list(ctx.args))
    config = Config.from_disk(config_path, overrides=overrides)
    nlp = init_nlp(config, device="cpu", sample_size=8)
    total = 0
    for name, pipe in nlp.pipeline:
        if component and name != component:
            continue
        if pipe.module is None:
            print(f"[{name}] (no parameters — rule component)")
            continue
        print(f"[{name}]")
        for pname, p in pipe.module.named_parameters():
            n = p.numel()
            total += n
            print(f"  {pname:48s} {str(tuple(p.shape)):>18s} {n:>10,d}")
    print(f"total parameters: {total:,d}")


@app.command("find-threshold")
def find_threshold_cli(
    model_path: Path = typer.Argument(..., help="Trained pipeline directory"),
    pipe_name: str = typer.Argument(..., help="Component to tune (spancat / textcat_multilabel)"),
    scores_key: str = typer.Option("", "--scores-key", help="Score to maximize (default: spans_sc_f for spancat, cats_macro_acc otherwise)"),
    n_trials: int = typer.Option(11, "--n-trials", help="Thresholds to try in (0, 1)"),
    use_gpu: int = typer.Option(-1, "--gpu-id", "-g"),
    corpus_dot: str = typer.Option("corpora.dev", "--corpus"),
):
    """Sweep a prediction threshold on dev data and report the best
    (spaCy's `find-threshold` role for spancat / multilabel textcat)."""
    import spacy_ray_amd
    from spacy_ray_amd.config.config import resolve_dot_names
    from spacy_ray_amd.train.scorer import score_examples

    device = f"cuda:{use_gpu}" if use_gpu >= 0 else "cpu"
    nlp = spacy_ray_amd.load(str(model_path), device=device)
    pipe = nlp.get_pipe(pipe_name)
    if not hasattr(pipe, "threshold"):
        raise SystemExit(f"{pipe_name} has no threshold to tune")
    icfg = nlp.config.interpolate()
    (corpus,) = resolve_dot_names(icfg, [corpus_dot])
    examples = list(corpus(nlp))
    if not examples:
        raise SystemExit(f"{corpus_dot} is empty")
    key = scores_key or ("spans_sc_f" if pipe_name == "spancat"
                         else "cats_macro_acc")
    best = (None, -1.0)
    for i in range(1, n_trials + 1):
        thr = i / (n_trials + 1)
        pipe.threshold = thr
        for eg in examples:
            eg.predicted = eg.predicted.copy_unannotated()
        nlp.predict_docs([eg.predicted for eg in examples])
        sc = score_examples(examples, [pipe_name]).get(key, 0.0)
        print(f"  threshold {thr:.3f}  {key} {sc:.4f}")
        if sc > best[1]:
            best = (thr, sc)
    print(f"[+] best threshold {best[0]:.3f} ({key} {best[1]:.4f}) — set "
          f"`threshold = {best[0]:.3f}` on [components.{pipe_name}]")


@app.command("pretrain")
def pretrain_cli(
    ctx: typer.Context,
    config_path: Path = typer.Argument(..., help="Path to config file (its tok2vec + train corpus are used)"),
    output_path: Path = typer.Argument(..., help="Output directory for tok2vec.safetensors"),
    steps: int = typer.Option(1000, "--steps", help="Pretraining steps"),
    batch_docs: int = typer.Option(64, "--batch-docs", help="Docs per step"),
    n_buckets: int = typer.Option(4096, "--buckets", help="Masked-token target buckets"),
    mask_rate: float = typer.Option(0.15, "--mask-rate"),
    learn_rate: float = typer.Option(1e-3, "--lr"),
    use_gpu: int = typer.Option(-1, "--gpu-id", "-g", help="GPU ID or -1 for CPU"),
):
    """Pretrain the config's tok2vec with a masked-token objective on the
    raw training corpus (the `spacy pretrain` role); load the result at
    training time with `--init-tok2vec` / `training.init_tok2vec`."""
    from spacy_ray_amd.config.config import resolve_dot_names
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.train.pretrain import pretrain_tok2vec, save_tok2vec

    overrides = parse_config_overrides(list(ctx.args))
    config = Config.from_disk(config_path, overrides=overrides)
    device = f"cuda:{use_gpu}" if use_gpu >= 0 else "cpu"
    nlp = init_nlp(config, device=device)
    icfg = config.interpolate()
    dot = icfg.get("training", {}).get("train_corpus", "corpora.train")
    (corpus,) = resolve_dot_names(icfg, [dot])
    losses = pretrain_tok2vec(nlp, corpus, steps=steps, batch_docs=batch_docs,
                              n_buckets=n_buckets, mask_rate=mask_rate,
                              learn_rate=learn_rate)
    path = save_tok2vec(nlp, output_path)
    print(f"[+] wrote {path} (final loss {losses[-1]:.4f}) — train with "
          f"--init-tok2vec {output_path}")


@app.command("package")
def package_cli(
    model_path: Path = typer.Argument(..., help="Trained pipeline directory (model-best/model-last)"),
    output_dir: Path = typer.Argument(..., help="Directory to create the package in"),
    name: str = typer.Option("pipeline", "--name", "-n", help="Package name (lang prefix added from meta)"),
    version: str = typer.Option("0.0.0", "--version", "-v"),
):
    """Wrap a trained pipeline into a pip-installable package (spaCy
    `package` role): <lang>_<name>-<version>/ with setup.py, a loader
    module and the model data."""
    import json
    import shutil

    meta = json.loads((model_path / "meta.json").read_text())
    lang = meta.get("lang", "xx")
    pkg = f"{lang}_{name}"
    root = output_dir / f"{pkg}-{version}"
    mod_dir = root / pkg
    data_dir = mod_dir / f"{pkg}-{version}"
    if data_dir.exists():
        shutil.rmtree(data_dir)
    data_dir.parent.mkdir(parents=True, exist_ok=True)
    shutil.copytree(model_path, data_dir)
    meta["name"] = name
    meta["version"] = version
    (data_dir / "meta.json").write_text(json.dumps(meta, indent=2))
    (mod_dir / "__init__.py").write_text(
        '"""Auto-generated pipeline package (spacy-mi package)."""\n'
        "from pathlib import Path\n\n"
        f"__version__ = {version!r}\n\n\n"
        "def load(device: str = \"cpu\"):\n"
        f"    from spacy_ray_amd import load as _load\n\n"
        f"    return _load(Path(__file__).parent / '{pkg}-{version}',"
        " device=device)\n"
    )
    (root / "setup.py").write_text(
        "from setuptools import setup\n\n"
        "setup(\n"
        f"    name={pkg!r},\n"
        f"    version={version!r},\n"
        f"    packages=[{pkg!r}],\n"
        "    include_package_data=True,\n"
        f"    package_data={{{pkg!r}: ['{pkg}-{version}/**/*', '{pkg}-{version}/*']}},\n"
        "    install_requires=['spacy_ray_amd'],\n"
        ")\n"
    )
    (root / "MANIFEST.in").write_text(f"recursive-include {pkg} *\n")
    print(f"package written -> {root}")


@app.command("serve")
def serve_cli(
    model_path: Path = typer.Argument(..., help="Trained pipeline directory (model-best/model-last)"),
    host: str = typer.Option("127.0.0.1", "--host"),
    port: int = typer.Option(8000, "--port", "-p"),
    use_gpu: int = typer.Option(-1, "--gpu-id", "-g", help="GPU ID or -1 for CPU"),
    max_batch: int = typer.Option(256, "--max-batch", help="Max docs per decode batch"),
    max_wait_ms: float = typer.Option(5.0, "--max-wait-ms", help="Micro-batcher wait for coalescing concurrent requests"),
):
    """Serve a trained pipeline over HTTP (POST /annotate, GET /info)."""
    from spacy_ray_amd.serve.app import serve

    serve(model_path, host=host, port=port, use_gpu=use_gpu,
          max_batch=max_batch, max_wait_ms=max_wait_ms)


def main() -> None:
    app()


if __name__ == "__main__":
    main()
