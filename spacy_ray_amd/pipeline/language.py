"""Language: the pipeline container (spaCy ``nlp`` contract).

Holds vocab + ordered pipes, runs the shared-tok2vec single-graph training
step (SURVEY.md §3.2 disposition), scores on dev data, and round-trips to a
spaCy-style on-disk layout: config.cfg + meta.json + vocab/ + one directory
per component with its params (`/root/reference/spacy_ray/worker.py:219-222`
``nlp.to_disk`` contract).
"""
from __future__ import annotations

import json
from pathlib import Path
from typing import Dict, List, Optional, Sequence, Tuple

import torch
import torch.nn as nn

from spacy_ray_amd.config.config import Config, resolve, resolve_dot_names
from spacy_ray_amd.config.registry import registry
from spacy_ray_amd.models.batch import TokenBatch
from spacy_ray_amd.vocab.doc import Doc, Example, Vocab
from .pipes import TrainablePipe, Tok2VecPipe


class Language:
    def __init__(self, vocab: Vocab, config: Config, device: str = "cpu") -> None:
        self.vocab = vocab
        self.config = config
        self.device = torch.device(device)
        self.pipeline: List[Tuple[str, TrainablePipe]] = []
        self.meta: Dict = {"lang": vocab.lang, "name": "pipeline", "version": "0.0.0"}
        self._frozen: List[str] = []
        self._annotating: List[str] = []
        # select_pipes(disable=...): skipped during inference AND frozen for
        # training (spaCy's disabled-component contract — ADVICE r1)
        self._disabled: List[str] = []
        # rule-based tokenizer (spaCy prefix/suffix/infix algorithm) for the
        # nlp(text) / pipe / serve surface; training from DocBin corpora
        # never tokenizes
        from spacy_ray_amd.vocab.tokenizer import Tokenizer

        self.tokenizer = Tokenizer()

    # ------------------------------------------------------------- pipeline
    @property
    def pipe_names(self) -> List[str]:
        return [n for n, _ in self.pipeline]

    def get_pipe(self, name: str) -> TrainablePipe:
        for n, p in self.pipeline:
            if n == name:
                return p
        raise KeyError(name)

    def add_pipe(self, name: str, pipe: TrainablePipe) -> None:
        self.pipeline.append((name, pipe))

    @property
    def tok2vec(self) -> Optional[Tok2VecPipe]:
        for _, p in self.pipeline:
            if isinstance(p, Tok2VecPipe):
                return p
        return None

    def torch_module(self) -> nn.ModuleDict:
        """All trainable parameters as one module (flat-buffer optimizer input).
        Iteration order == pipeline order == identical on every rank (the
        cross-rank key invariant)."""
        d = nn.ModuleDict()
        for name, pipe in self.pipeline:
            if pipe.module is not None:
                d[name.replace(".", "_")] = pipe.module
        return d

    # ------------------------------------------------------------- training
    def forward_loss(self, examples: Sequence[Example], losses: Optional[Dict[str, float]] = None,
                     drop: float = 0.0, token_batch: Optional[TokenBatch] = None):
        """One forward pass over all trainable pipes -> (total_loss, losses).

        `token_batch`: optional precomputed TokenBatch for these examples —
        callers replaying fixed batches (bench) skip the per-step
        concat+upload."""
        from spacy_ray_amd.utils import timing

        if losses is None:
            losses = {}
        with timing.phase("data/batch_build"):
            batch = token_batch if token_batch is not None else TokenBatch(
                [eg.predicted for eg in examples], self.device)
        from .pipes import _TransitionPipeBase, run_transition_tasks

        # gold staging FIRST (empty GPU queue): pageable H2D after kernels
        # are queued blocks the host behind all of them
        for name, pipe in self.pipeline:
            if name not in self._frozen and hasattr(pipe, "stage_gold"):
                pipe.stage_gold(examples, batch)
        t2v_pipe = self.tok2vec
        with timing.phase("fwd/tok2vec"):
            t2v = t2v_pipe.forward(batch, drop=drop) if t2v_pipe is not None else None
        if t2v is not None and t2v_pipe.name in self._frozen:
            # frozen tok2vec: listeners may read it but must not update it
            t2v = t2v.detach()
        if self._annotating:
            # training.annotating_components (spaCy contract): the listed
            # pipes SET their predictions on eg.predicted before any loss is
            # computed, so downstream components can read the annotations.
            # No built-in pipe consumes another's annotations today, but the
            # mechanism is live for user components (--code factories).
            self._run_annotating(examples, batch, t2v)
        total = None
        # transition pipes (parser/NER) run INTERLEAVED: their per-step
        # GPU-score / CPU-advance phases pipeline against each other
        trans_tasks = []
        for name, pipe in self.pipeline:
            if (isinstance(pipe, Tok2VecPipe) or name in self._frozen
                    or not getattr(pipe, "trainable", True)):
                continue
            own = pipe.own_tok2vec(batch, drop=drop)
            pt2v = own if own is not None else t2v
            if isinstance(pipe, _TransitionPipeBase):
                with timing.phase(f"loss/make_task_{name}"):
                    trans_tasks.append((name, pipe,
                                        pipe.make_loss_task(examples, pt2v, batch)))
                continue
            with timing.phase(f"loss/{name}"):
                loss, display = pipe.get_loss(examples, pt2v, batch)
            losses[name] = losses.get(name, 0.0) + display
            total = loss if total is None else total + loss
        if trans_tasks:
            with timing.phase("loss/transition_steps"):
                run_transition_tasks([t for _, _, t in trans_tasks])
            for name, pipe, task in trans_tasks:
                with timing.phase(f"loss/{name}"):
                    loss, display = pipe.finish_task(task)
                losses[name] = losses.get(name, 0.0) + display
                total = loss if total is None else total + loss
        if total is None:
            total = torch.zeros((), device=self.device)
        return total, losses

    def _run_annotating(self, examples, batch, t2v) -> None:
        from .pipes import _TransitionPipeBase, run_transition_tasks

        docs = [eg.predicted for eg in examples]
        with torch.no_grad():
            t2v_d = t2v.detach() if t2v is not None else None
            for name, pipe in self.pipeline:
                if name not in self._annotating or isinstance(pipe, Tok2VecPipe):
                    continue
                own = pipe.own_tok2vec(batch)
                pt2v = own if own is not None else t2v_d
                if (isinstance(pipe, _TransitionPipeBase)
                        and getattr(pipe, "beam_width", 1) <= 1):
                    task, splits, shards = pipe.make_predict_task(docs, pt2v)
                    run_transition_tasks([task])
                    if task.gpu is not None:  # GPU state machine decode
                        pipe._annotate_gpu(docs, task.gpu_decode)
                        continue
                    for (lo, hi, base), states in zip(splits, shards):
                        pipe._annotate(docs[lo:hi], states)
                else:
                    pipe.predict_and_set(docs, pt2v, batch)

    def update(self, examples: Sequence[Example], *, sgd=None, losses=None, drop: float = 0.0):
        """Convenience single-process update (tests / CPU smoke)."""
        total, losses = self.forward_loss(examples, losses, drop=drop)
        total.backward()
        if sgd is not None:
            sgd.step()
            sgd.zero_grad()
        for k, v in losses.items():  # deferred display-loss tensors
            if torch.is_tensor(v):
                losses[k] = float(v)
        return losses

    # ------------------------------------------------------------ inference
    def predict_docs(self, docs: Sequence[Doc]) -> Sequence[Doc]:
        from .pipes import _TransitionPipeBase, run_transition_tasks

        if not docs:
            return docs
        with torch.no_grad():
            batch = TokenBatch(docs, self.device)
            t2v_pipe = self.tok2vec
            t2v = t2v_pipe.forward(batch) if t2v_pipe is not None else None
            # interleave the transition pipes' decode loops
            trans = []
            for name, pipe in self.pipeline:
                if name in self._disabled:
                    continue
                if (isinstance(pipe, _TransitionPipeBase)
                        and getattr(pipe, "beam_width", 1) <= 1):
                    own = pipe.own_tok2vec(batch)
                    trans.append((pipe,) + pipe.make_predict_task(
                        docs, own if own is not None else t2v))
            if trans:
                run_transition_tasks([t[1] for t in trans])
            heads = []
            for name, pipe in self.pipeline:
                if (isinstance(pipe, Tok2VecPipe) or name in self._disabled
                        or (isinstance(pipe, _TransitionPipeBase)
                            and getattr(pipe, "beam_width", 1) <= 1)):
                    continue
                own = pipe.own_tok2vec(batch)
                heads.append((name, pipe, own if own is not None else t2v))
        # annotations land in PIPELINE ORDER (spaCy contract): a later
        # component sees / may respect an earlier one's output — e.g.
        # entity_ruler BEFORE ner keeps ruler entities only if ner runs
        # later by position, ner-then-entity_ruler lets the ruler fill O
        # tokens.  (The decode LOOPS above still ran interleaved; only the
        # doc writes are ordered here.)
        head_by_name = {name: (pipe, pt2v) for name, pipe, pt2v in heads}
        trans_by_name = {pipe.name: (pipe, task, splits, shards)
                         for pipe, task, splits, shards in trans}
        for name, _ in self.pipeline:
            if name in head_by_name:
                pipe, pt2v = head_by_name[name]
                pipe.predict_and_set(docs, pt2v, batch)
            elif name in trans_by_name:
                pipe, task, splits, shards = trans_by_name[name]
                if task.gpu is not None:  # GPU state machine: decode tensors
                    pipe._annotate_gpu(docs, task.gpu_decode)
                    continue
                for (lo, hi, base), states in zip(splits, shards):
                    pipe._annotate(docs[lo:hi], states)
        return docs

    def evaluate(self, examples: Sequence[Example], batch_size: int = 256) -> Dict[str, float]:
        from spacy_ray_amd.train.scorer import score_examples

        for i in range(0, len(examples), batch_size):
            chunk = examples[i : i + batch_size]
            self.predict_docs([eg.predicted for eg in chunk])
        return score_examples(examples, self.pipe_names)

    def pipe(self, inputs, batch_size: int = 256):
        """Annotate a stream of Docs or texts in batches (spaCy nlp.pipe
        contract)."""
        batch: List[Doc] = []
        for item in inputs:
            doc = item if isinstance(item, Doc) else self.tokenizer(self.vocab, item)
            batch.append(doc)
            if len(batch) >= batch_size:
                yield from self.predict_docs(batch)
                batch = []
        if batch:
            yield from self.predict_docs(batch)

    def __call__(self, text: str) -> Doc:
        doc = self.tokenizer(self.vocab, text)
        self.predict_docs([doc])
        return doc

    # ---------------------------------------------------------- persistence
    def to_disk(self, path) -> None:
        path = Path(path)
        path.mkdir(parents=True, exist_ok=True)
        self.config.to_disk(path / "config.cfg")
        # spaCy meta shape: pipeline order + per-component labels
        self.meta["pipeline"] = self.pipe_names
        self.meta["labels"] = {
            name: list(getattr(pipe, "labels", []) or [])
            for name, pipe in self.pipeline
        }
        (path / "meta.json").write_text(json.dumps(self.meta, indent=2))
        vocab_dir = path / "vocab"
        vocab_dir.mkdir(exist_ok=True)
        (vocab_dir / "strings.json").write_text(json.dumps(self.vocab.strings.to_list()))
        # tokenizer settings file (spaCy layout has a `tokenizer` blob)
        from spacy_ray_amd.vocab.tokenizer import tokenizer_to_bytes

        (path / "tokenizer").write_bytes(tokenizer_to_bytes(getattr(self, "tokenizer", None)))
        for name, pipe in self.pipeline:
            pdir = path / name
            pdir.mkdir(exist_ok=True)
            (pdir / "cfg.json").write_text(json.dumps(pipe.state_cfg()))
            if pipe.module is not None:
                from safetensors.torch import save_file

                state = {k: v.detach().cpu().contiguous() for k, v in pipe.module.state_dict().items()}
                save_file(state, str(pdir / "model.safetensors"))
                # Thinc-msgpack `model` bytes alongside (spaCy component
                # layout: <component>/model; see data/thinc_serde.py)
                from spacy_ray_amd.data.thinc_serde import (
                    component_to_thinc_nodes, model_to_thinc_bytes)

                (pdir / "model").write_bytes(
                    model_to_thinc_bytes(component_to_thinc_nodes(pipe)))

    def from_disk(self, path) -> "Language":
        path = Path(path)
        self.meta = json.loads((path / "meta.json").read_text())
        if (path / "tokenizer").exists():
            from spacy_ray_amd.vocab.tokenizer import tokenizer_from_bytes

            self.tokenizer = tokenizer_from_bytes((path / "tokenizer").read_bytes())
        strings = json.loads((path / "vocab" / "strings.json").read_text())
        for s in strings:
            self.vocab.strings.add(s)
        for name, pipe in self.pipeline:
            pdir = path / name
            if (pdir / "cfg.json").exists():
                pipe.load_cfg(json.loads((pdir / "cfg.json").read_text()), self.device)
            mfile = pdir / "model.safetensors"
            if mfile.exists() and pipe.module is not None:
                from safetensors.torch import load_file

                state = load_file(str(mfile))
                pipe.module.load_state_dict(state)
                pipe.module.to(self.device)
            elif (pdir / "model").exists() and pipe.module is not None:
                # spaCy-written checkpoint: Thinc-msgpack component bytes
                from spacy_ray_amd.data.thinc_serde import (
                    load_thinc_nodes_into_component)

                n = load_thinc_nodes_into_component(
                    pipe, (pdir / "model").read_bytes())
                if n == 0:
                    raise ValueError(
                        f"component {name!r}: no tensors from the Thinc "
                        f"model bytes matched this architecture"
                    )
                pipe.module.to(self.device)
        return self

    def select_pipes(self, disable: Sequence[str]):
        """Disabled pipes are skipped during inference (predict_docs /
        __call__ / pipe) and frozen for training, matching spaCy's
        select_pipes contract."""
        lang = self

        class _Ctx:
            def __enter__(self):
                lang._frozen = list(disable)
                lang._disabled = list(disable)
                return lang

            def __exit__(self, *a):
                lang._frozen = []
                lang._disabled = []

        return _Ctx()


def build_nlp(config: Config, device: str = "cpu") -> Language:
    """Build the pipeline skeleton from [nlp]/[components] (uninitialized)."""
    registry.ensure_populated()
    cfg = config.interpolate()
    lang = cfg.get("nlp", {}).get("lang", "xx")
    pipe_names = cfg.get("nlp", {}).get("pipeline", [])
    nlp = Language(Vocab(lang), config, device=device)
    components = cfg.get("components", {})
    for name in pipe_names:
        comp_cfg = dict(components.get(name, {}))
        factory_name = comp_cfg.pop("factory", name)
        model_cfg = comp_cfg.pop("model", None)
        model_spec = resolve(model_cfg) if model_cfg else None
        factory = registry.factories.get(factory_name)
        pipe = factory(name=name, model=model_spec, **comp_cfg)
        nlp.add_pipe(name, pipe)
    _check_listener_widths(nlp)
    return nlp


def _check_listener_widths(nlp: Language) -> None:
    """Fail at build time (not deep inside a matmul) when a listener head's
    width disagrees with the shared encoder, or when a listener has no
    shared encoder to listen to."""
    shared = nlp.tok2vec
    for name, pipe in nlp.pipeline:
        if isinstance(pipe, Tok2VecPipe) or pipe.listens_to is None:
            continue
        if getattr(pipe, "embedded_spec", None) is not None:
            continue  # owns its encoder
        if shared is None:
            raise ValueError(
                f"component {name!r} listens to a shared tok2vec but the "
                f"pipeline {nlp.pipe_names} has none — add a tok2vec pipe or "
                f"give {name!r} an embedded [components.{name}.model.tok2vec] "
                f"block (a full architecture, not a Tok2VecListener)"
            )
        if pipe.width != shared.width:
            raise ValueError(
                f"component {name!r} expects width {pipe.width} but the "
                f"shared tok2vec produces width {shared.width} — align "
                f"[components.{name}.model.tok2vec.width] with the encoder"
            )


def init_nlp(config: Config, device: str = "cpu", sample_size: int = 128) -> Language:
    """Build + initialize (label discovery + param init) — the contract of
    spaCy's init_nlp at `/root/reference/spacy_ray/worker.py:91`.
    Deterministic: seeds torch RNG from [training.seed] BEFORE building so
    every rank gets identical initial params (SURVEY.md §3.4 invariant).

    Label discovery runs over the FULL training corpus (one epoch), like
    spaCy's init-labels pass — a label first appearing late in the corpus
    must not crash mid-training.  Components may instead pin labels
    explicitly via `labels = [...]` in their [components.*] block, which
    skips discovery for that pipe.  Module shape-init still uses only the
    first `sample_size` examples."""
    cfg = config.interpolate()
    seed = int(cfg.get("training", {}).get("seed", 0) or 0)
    torch.manual_seed(seed)
    nlp = build_nlp(config, device=device)
    (train_corpus,) = resolve_dot_names(cfg, [cfg["training"]["train_corpus"]])
    if "transformer" in nlp.pipe_names:
        # the transformer trains its byte-level BPE on this sample at init;
        # a tiny sample under-merges and every word fragments into many
        # subwords (3-4x sequence inflation = 3-4x activation memory)
        sample_size = max(sample_size, 1024)
    sample: List[Example] = []
    # generic full-corpus discovery: each pipe declares what it reads from
    # a reference Doc (labels_from) — subclass-safe (morphologizer IS a
    # TaggerPipe but reads morphs, senter has fixed labels)
    discover = [
        (n, p) for n, p in nlp.pipeline
        if not getattr(p, "labels", None) and hasattr(p, "label2id")
        and hasattr(p, "labels_from")
    ]
    label_sets: Dict[str, set] = {n: set() for n, _ in discover}
    for eg in train_corpus(nlp):
        if len(sample) < sample_size:
            sample.append(eg)
        if not discover:
            if len(sample) >= sample_size:
                break
            continue
        ref = eg.reference
        for n, p in discover:
            label_sets[n].update(p.labels_from(ref))
    for name, pipe in discover:
        if label_sets[name]:
            pipe.labels = sorted(label_sets[name])
        pipe.label2id = {t: i for i, t in enumerate(pipe.labels)}
    for name, pipe in nlp.pipeline:
        pipe.initialize(sample, nlp.device)
    return nlp
