"""Trainable pipes: tok2vec, tagger, transition parser, NER.

Re-design of spaCy's pipeline components for a single-autograd-graph step
(SURVEY.md §3.2 disposition): the tok2vec pipe runs ONCE per batch; tagger/
parser/NER are listeners consuming the shared [T, W] tensor; all losses sum
into one backward so the distributed engine sees a single gradient pass to
overlap with.
"""
from __future__ import annotations

import itertools
import os
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch
import torch.nn as nn

from spacy_ray_amd import _srx_cpu
from spacy_ray_amd.models.batch import TokenBatch
from spacy_ray_amd.models.parser_model import TransitionModel
from spacy_ray_amd.vocab.doc import Example, biluo_to_codes

NEG_INF = -1e30


class TrainablePipe:
    name: str
    listens_to: Optional[str] = None
    trainable: bool = True  # rule-based pipes (entity_ruler) set False

    def __init__(self) -> None:
        self.module: Optional[nn.Module] = None
        self.cfg: Dict = {}

    def initialize(self, examples: Sequence[Example], device) -> None:
        raise NotImplementedError

    def get_loss(self, examples, t2v, batch) -> Tuple[torch.Tensor, float]:
        """-> (loss tensor for the main backward, display float for logging).

        The tensor may be a SURROGATE whose value is meaningless but whose
        gradient is exact (transition pipes); always log the display float."""
        raise NotImplementedError

    def predict_and_set(self, docs, t2v, batch) -> None:
        raise NotImplementedError

    # ------- serialization hooks (labels etc.); params live in module
    def state_cfg(self) -> Dict:
        return dict(self.cfg)

    def load_cfg(self, cfg: Dict, device) -> None:
        self.cfg = dict(cfg)

    # ------- embedded (non-listener) tok2vec: a pipe whose config carries a
    # full tok2vec block owns its encoder (spaCy's per-component tok2vec).
    # The encoder is registered as a submodule of the head module so the
    # flat-buffer engine and checkpoints see its params automatically.
    embedded_spec = None  # set by subclasses from their ModelSpec

    def _attach_embedded(self, device) -> None:
        if (self.embedded_spec is not None and self.module is not None
                and not hasattr(self.module, "embedded_t2v")):
            self.module.embedded_t2v = self.embedded_spec.build().to(device)

    def own_tok2vec(self, batch, drop: float = 0.0):
        """This pipe's private tok2vec output, or None if it listens to the
        shared encoder."""
        emb = getattr(self.module, "embedded_t2v", None) if self.module is not None else None
        return None if emb is None else emb(batch, drop=drop)


class Tok2VecPipe(TrainablePipe):
    name = "tok2vec"

    def __init__(self, name: str, spec) -> None:
        super().__init__()
        self.name = name
        self.spec = spec
        self.width = spec.width

    def initialize(self, examples, device) -> None:
        if self.module is None:
            self.module = self.spec.build().to(device)
        if hasattr(self.module, "init_bpe"):
            # transformer tok2vec: train/load the byte-level BPE on the
            # corpus sample (offline — no pretrained vocab available)
            self.module.init_bpe(examples)

    def load_cfg(self, cfg, device) -> None:
        super().load_cfg(cfg, device)
        if self.module is None:
            self.module = self.spec.build().to(device)
        if hasattr(self.module, "init_bpe") and "bpe_vocab" in cfg:
            # checkpoint restore: rebuild the trained tokenizer from its
            # serialized JSON (saved in state_cfg)
            from tokenizers import Tokenizer as HFTokenizer

            self.module.bpe.tok = HFTokenizer.from_str(cfg["bpe_vocab"])

    def forward(self, batch: TokenBatch, drop: float = 0.0) -> torch.Tensor:
        return self.module(batch, drop=drop)

    def state_cfg(self) -> Dict:
        cfg = dict(self.cfg)
        bpe = getattr(self.module, "bpe", None) if self.module is not None else None
        if bpe is not None and bpe.tok is not None:
            cfg["bpe_vocab"] = bpe.tok.to_str()  # tokenizer.json content
        return cfg

    def get_loss(self, examples, t2v, batch):
        return t2v.new_zeros(()), 0.0

    def predict_and_set(self, docs, t2v, batch) -> None:
        pass  # downstream pipes consume t2v directly


class TaggerHead(nn.Module):
    def __init__(self, width: int, n_tags: int):
        super().__init__()
        self.output = nn.Linear(width, n_tags)
        nn.init.zeros_(self.output.weight)
        nn.init.zeros_(self.output.bias)

    def forward(self, X):
        from spacy_ray_amd.ops.api import linear_cdw

        return linear_cdw(X, self.output.weight, self.output.bias)


class TaggerPipe(TrainablePipe):
    name = "tagger"
    listens_to = "tok2vec"

    @staticmethod
    def labels_from(ref):
        """Labels this pipe reads from a reference Doc (init_nlp's
        full-corpus discovery hook; subclasses override)."""
        return ref.tags or ()

    def __init__(self, name: str, spec) -> None:
        super().__init__()
        self.name = name
        self.width = spec.width
        self.embedded_spec = getattr(spec, "embedded_tok2vec", None)
        self.labels: List[str] = []
        self.label2id: Dict[str, int] = {}

    def initialize(self, examples, device) -> None:
        if not self.labels:
            labels = set()
            for eg in examples:
                if eg.reference.tags:
                    labels.update(eg.reference.tags)
            self.labels = sorted(labels)
            self.label2id = {t: i for i, t in enumerate(self.labels)}
        self.cfg["labels"] = self.labels
        if self.module is None:
            self.module = TaggerHead(self.width, max(1, len(self.labels))).to(device)
        self._attach_embedded(device)

    def load_cfg(self, cfg, device) -> None:
        super().load_cfg(cfg, device)
        self.labels = list(cfg.get("labels", []))
        self.label2id = {t: i for i, t in enumerate(self.labels)}
        n = max(1, len(self.labels))
        if self.module is None:
            # keep an existing module: its params may already be views into
            # the distributed engine's flat buffer (resume path)
            self.module = TaggerHead(self.width, n).to(device)
        elif self.module.output.out_features != n:
            raise ValueError(
                f"{self.name}: checkpoint has {n} labels but the initialized "
                f"module has {self.module.output.out_features}"
            )
        self._attach_embedded(device)

    def _gold_ids(self, examples, n_tokens: int) -> np.ndarray:
        # per-doc gold-id arrays are cached on the reference Doc (dict
        # lookups over every token measured ~50 ms/step at 512k words;
        # corpora recycle docs across epochs so the cache pays)
        # keyed by the label LIST identity-proof tuple, not id() — a
        # freed dict's address can be reused by a rebuilt pipeline and
        # would hit stale cached ids on shared docs
        key = ("tag_ids", tuple(self.labels))
        parts = []
        for eg in examples:
            ref = eg.reference
            cached = ref.user_data.get(key)
            if cached is None:
                tags = ref.tags or ["" for _ in range(len(ref))]
                cached = np.fromiter(
                    (self.label2id.get(t, -1) for t in tags), dtype=np.int64,
                    count=len(tags),
                )
                ref.user_data[key] = cached
            parts.append(cached)
        ids = np.concatenate(parts) if parts else np.zeros(0, dtype=np.int64)
        if len(ids) < n_tokens:  # batch pad rows: ignore_index
            ids = np.concatenate([ids, np.full(n_tokens - len(ids), -1, dtype=np.int64)])
        return ids

    def stage_gold(self, examples, batch) -> None:
        """Upload gold ids at STEP START (empty GPU queue): a pageable H2D
        later in the step blocks the host behind every queued kernel
        (measured ~56 ms/step at 1M words).  Cached on the TokenBatch for
        replayed batches."""
        key = ("tagger_gold", self.name)
        if batch is None or key in batch.staged:
            return
        from spacy_ray_amd.utils.pinned import to_device

        gold_np = self._gold_ids(examples, batch.n_tokens)
        gold = to_device(gold_np, batch.attr_ids.device)
        batch.staged[key] = (gold, int((gold_np >= 0).sum()))

    def get_loss(self, examples, t2v, batch):
        from spacy_ray_amd.ops import api as _ops

        scores = self.module(t2v)  # [T, nT]
        staged = batch.staged.get(("tagger_gold", self.name)) if batch is not None else None
        if staged is not None and staged[0].shape[0] == t2v.shape[0]:
            gold, n = staged
        else:
            from spacy_ray_amd.utils.pinned import to_device

            gold_np = self._gold_ids(examples, t2v.shape[0])
            gold = to_device(gold_np, scores.device)
            n = int((gold_np >= 0).sum())
        loss = _ops.softmax_ce_loss(scores, gold) / max(1, n)
        # display is a detached 0-dim tensor: converting to float here would
        # device-sync mid-forward; the engine converts once after backward
        return loss, loss.detach()

    def predict_and_set(self, docs, t2v, batch) -> None:
        with torch.no_grad():
            pred = self.module(t2v).argmax(dim=-1).cpu().numpy()
        # vectorized id->tag lookup over the whole batch
        table = (np.asarray(self.labels, dtype=object) if self.labels
                 else np.asarray([""], dtype=object))
        tag_strs = table[np.clip(pred, 0, len(table) - 1)].tolist()
        off = 0
        for doc in docs:
            n = len(doc)
            doc.tags = tag_strs[off:off + n]
            off += n


_next_task_id = itertools.count(1).__next__


class _TransitionTask:
    """One pipe's transition-loop context: states shards, the detached
    precompute tensor, the persistent dPre gradient accumulator and the
    per-step loss collections.

    Two levels of overlap engineering (SURVEY.md §2.2 N8 disposition):
    1. TWO-PHASE backward: per-step losses (upper GEMM + bias) are
       backpropagated in finish_task while each step's dPre scatters into
       ONE persistent fp32 buffer (ops.parser_step_score_accum); the loss
       returned upward is a zero-valued gradient-injection node
       (ops.inject_grad) that hands exactly dPre to the precompute GEMM in
       the caller's single main backward.
    2. TASK/SHARD PIPELINING (run_transition_tasks): while one unit's
       actions travel GPU->CPU and its C++ state machine advances, another
       unit's scoring kernels run — parser and NER loops interleave so the
       GPU never idles on a single pipe's sync."""

    __slots__ = ("pipe", "shards", "t2v", "train", "T", "pre", "pre_d",
                 "dPre32", "hip", "score_chunks", "gold_chunks",
                 "valid_chunks", "n_states_total", "entries", "fused",
                 "task_id", "cpp", "cpp_outs", "gpu", "gpu_decode",
                 "steps_dev")

    def __init__(self, pipe, shards, t2v, train: bool) -> None:
        from spacy_ray_amd.ops import api as _ops

        self.pipe = pipe
        self.shards = shards
        self.t2v = t2v
        self.train = train
        device = t2v.device
        self.T = t2v.shape[0]
        self.pre = pipe.module.precompute(t2v)
        self.pre_d = self.pre.detach()
        self.hip = _ops.hip_ext() if device.type == "cuda" else None
        # C++-owned loop (srx_steploop.hip): ONE python call runs the whole
        # per-batch transition loop; arenas come back for the batched loss.
        # SRX_CPP_LOOP=0 falls back to the python round-robin (parity tests).
        self.cpp = (
            self.hip is not None
            and hasattr(self.hip, "run_transition_loop")
            and os.environ.get("SRX_CPP_LOOP", "1") == "1"
        )
        self.cpp_outs: List = []
        self.dPre32 = (
            torch.zeros(self.pre.shape, dtype=torch.float32, device=device)
            if (train and not self.cpp) else None
        )
        # GPU train: per-step backwards append (feats, dSummed) here; the
        # scatter happens once, batched+sorted, in finish_task
        self.entries: Optional[List] = [] if (train and self.hip is not None) else None
        # fully-fused C++ step (scorer kernel + upper GEMM + action select in
        # ONE python call, backward entries stashed in a C++ store); default
        # on — GPU-validated vs the python path (test_fused_step_parity);
        # SRX_FUSED_STEP=0 falls back.  See fusedstep in srx_ext.hip.
        self.fused = (
            self.hip is not None
            and hasattr(self.hip, "fused_step")
            and os.environ.get("SRX_FUSED_STEP", "1") == "1"
        )
        self.task_id = _next_task_id() if self.fused else 0
        self.score_chunks: List[torch.Tensor] = []
        self.gold_chunks: List[torch.Tensor] = []
        self.valid_chunks: List[torch.Tensor] = []
        self.n_states_total = 0
        # GPU-resident state machine (srx_gpustate.hip): set by
        # make_loss_task/make_predict_task to the device-gold dict; the
        # whole transition loop then runs in ONE kernel launch.
        self.gpu: Optional[Dict] = None
        self.gpu_decode = None
        self.steps_dev: Optional[torch.Tensor] = None

    def launch(self, states):
        from spacy_ray_amd.ops import api as _ops
        from spacy_ray_amd.utils import timing
        from spacy_ray_amd.utils.pinned import to_device

        device = self.t2v.device
        with timing.span("raw/states_cpu"):
            # ONE packed buffer = [feats int64 | valid u8 | gold u8]; feats
            # missing slots already remapped to the pad row (= T, the
            # learned-pad index in precompute); ONE H2D copy per step
            act_idx, packed, nF = states.step_arrays_packed(self.train, self.T)
        Sa = len(act_idx)
        if Sa == 0:
            return None
        A = states.n_actions
        fbytes = Sa * nF * 8
        with timing.span("raw/score_fwd"):
            dev = to_device(packed, device)
            feats_t = dev[:fbytes].view(torch.int64).view(Sa, nF)
            valid_t = dev[fbytes:fbytes + Sa * A].view(Sa, A)
            if self.fused and self.train:
                gold_t = dev[fbytes + Sa * A:].view(Sa, A)
                mod = self.pipe.module
                scores, actions_dev = self.hip.fused_step(
                    self.pre_d, feats_t, mod.lower_b, mod.upper.weight,
                    mod.upper.bias, gold_t, valid_t, self.task_id, True,
                )
                self.score_chunks.append(scores)
                self.gold_chunks.append(gold_t)
                self.valid_chunks.append(valid_t)
                return act_idx, actions_dev, None
            if self.fused:
                # decode: same fused call under no_grad (no autograd node ->
                # nothing lands in the C++ entries store); valid doubles as
                # the "gold" mask so argmax-over-valid is selected
                mod = self.pipe.module
                with torch.no_grad():
                    _, actions_dev = self.hip.fused_step(
                        self.pre_d, feats_t, mod.lower_b, mod.upper.weight,
                        mod.upper.bias, valid_t, valid_t, 0, False,
                    )
                return act_idx, actions_dev, None
            if self.train:
                hidden = _ops.parser_step_score_accum(
                    self.pre_d, feats_t, self.pipe.module.lower_b,
                    self.dPre32, self.entries,
                )
                scores = self.pipe.module.upper(hidden)  # [S_active, A]
            else:
                scores = self.pipe.module.score(self.pre_d, feats_t)
        if self.train:
            with timing.span("raw/loss_build"):
                gold_t = dev[fbytes + Sa * A:].view(Sa, A)
                self.score_chunks.append(scores)
                self.gold_chunks.append(gold_t)
                self.valid_chunks.append(valid_t)
            sel_gold, sel_valid = gold_t, valid_t
        else:
            sel_gold, sel_valid = valid_t, valid_t
        if self.hip is not None:
            actions_dev = self.hip.action_select(scores.detach(), sel_gold, sel_valid)
            return act_idx, actions_dev, None
        s_np = scores.detach().float().cpu().numpy()
        valid_np = packed[fbytes:fbytes + Sa * A].reshape(Sa, A)
        g_np = (packed[fbytes + Sa * A:].reshape(Sa, A) if self.train else valid_np)
        choose = np.where(g_np > 0, s_np, NEG_INF)
        fallback = np.where(valid_np > 0, s_np, NEG_INF)
        any_gold = (g_np > 0).any(axis=1, keepdims=True)
        choose = np.where(any_gold, choose, fallback)
        return act_idx, None, choose.argmax(axis=1).astype(np.int32)

    def complete(self, states, pend) -> None:
        from spacy_ray_amd.utils import timing

        act_idx, actions_dev, actions_np = pend
        with timing.span("raw/score_d2h"):
            if actions_dev is not None:
                actions_np = actions_dev.cpu().numpy().astype(np.int32)
        with timing.span("raw/advance_cpu"):
            actions = np.full(len(states), -1, dtype=np.int32)
            actions[act_idx] = actions_np
            states.advance(actions)
        if self.train:
            self.n_states_total += len(act_idx)


def run_transition_tasks(tasks: List[_TransitionTask]) -> None:
    """Interleaved driver over every (task, shard) unit: complete the
    pending step (sync + C++ advance) then launch the next, round-robin —
    one unit's CPU work hides under another unit's GPU work.

    GPU default: the whole loop runs in C++ (hip.run_transition_loop, ONE
    python crossing per batch); the python round-robin below is the CPU /
    SRX_CPP_LOOP=0 fallback."""
    # GPU-resident state machines first: each is ONE async kernel launch on
    # the current stream, so any remaining host-loop tasks below overlap it.
    gpu_tasks = [t for t in tasks if t.gpu is not None]
    if gpu_tasks:
        from spacy_ray_amd.utils import timing

        with timing.span("raw/gpu_state"):
            for t in gpu_tasks:
                t.pipe._gpu_launch(t)
        tasks = [t for t in tasks if t.gpu is None]
    units = [(t, s) for t in tasks for s in t.shards]
    if not units:
        return
    if all(t.cpp for t in tasks):
        from spacy_ray_amd.utils import timing

        args = []
        for t, s in units:
            mod = t.pipe.module
            args.append((
                s.handle(), t.pre_d.contiguous(), mod.lower_b.detach(),
                mod.upper.weight.detach(), mod.upper.bias.detach(), t.train,
            ))
        with timing.span("raw/cpp_loop"):
            outs = tasks[0].hip.run_transition_loop(args)
        for (t, s), o in zip(units, outs):
            if t.train:
                t.cpp_outs.append(o)
        return
    pend = [None] * len(units)
    done = [False] * len(units)
    max_steps = max(4 * t.T + 16 for t in tasks)
    for _ in range(max_steps):
        progressed = False
        for k, (task, states) in enumerate(units):
            if done[k]:
                continue
            if pend[k] is not None:
                task.complete(states, pend[k])
                pend[k] = None
            out = task.launch(states)
            if out is None:
                done[k] = True
                continue
            pend[k] = out
            progressed = True
        if not progressed:
            break
    for k, (task, states) in enumerate(units):  # drain
        if pend[k] is not None:
            task.complete(states, pend[k])
            pend[k] = None


class _TransitionPipeBase(TrainablePipe):
    """Shared greedy transition machinery for parser and NER.

    Per step the C++ batch object returns ONE packed buffer (features +
    valid + min-cost masks for the ACTIVE states); the GPU scores them
    (fused gather+maxout + upper GEMM) and selects actions on-device;
    training follows the best-scoring min-cost action; the loss is one CE
    of softmax-over-valid against the uniform min-cost target computed over
    all steps at once (contract of spaCy's parser loss).
    See _TransitionTask for the two-phase backward and the task/shard
    pipelining."""

    listens_to = "tok2vec"

    def __init__(self, name: str, spec) -> None:
        super().__init__()
        self.name = name
        self.spec = spec
        self.width = spec.width
        self.embedded_spec = getattr(spec, "embedded_tok2vec", None)
        self.hidden_width = getattr(spec, "hidden_width", 64)
        self.nF = getattr(spec, "nF", 13)
        self.labels: List[str] = []
        self.label2id: Dict[str, int] = {}

    def _build_module(self, device) -> None:
        if self.module is None:
            self.module = TransitionModel(
                self.width, hidden_width=self.hidden_width, nF=self.nF
            ).to(device)
            self.module.initialize_output(self._n_actions())
            self.module.to(device)
        self._attach_embedded(device)

    def load_cfg(self, cfg, device) -> None:
        super().load_cfg(cfg, device)
        self.labels = list(cfg.get("labels", []))
        self.label2id = {t: i for i, t in enumerate(self.labels)}
        self._build_module(device)

    # ---- subclass hooks
    def _n_actions(self) -> int:
        raise NotImplementedError

    def _make_states(self, lengths: np.ndarray, base: int = 0):
        raise NotImplementedError

    def _set_gold(self, states, staged) -> None:
        raise NotImplementedError

    def _annotate(self, docs, states) -> None:
        raise NotImplementedError

    # ---- shared machinery
    def _split_docs(self, lengths: np.ndarray, n_shards: int):
        """Split docs into ~word-balanced contiguous shards:
        [(doc_lo, doc_hi, token_base), ...]."""
        if n_shards <= 1 or len(lengths) < 2 * n_shards:
            return [(0, len(lengths), 0)]
        csum = np.concatenate([[0], np.cumsum(lengths)])
        total = csum[-1]
        out = []
        lo = 0
        for k in range(n_shards):
            target = total * (k + 1) / n_shards
            hi = int(np.searchsorted(csum, target)) if k < n_shards - 1 else len(lengths)
            hi = max(hi, lo + 1)
            out.append((lo, hi, int(csum[lo])))
            lo = hi
            if lo >= len(lengths):
                break
        return out

    def begin_task(self, shards, t2v, train: bool) -> "_TransitionTask":
        """Build a step-loop task for this pipe (states + precompute +
        gradient accumulator + loss collections)."""
        return _TransitionTask(self, shards, t2v, train)

    def finish_task(self, task: "_TransitionTask"):
        """Compute the one-shot CE over all collected steps, run the phase-1
        backward, return (surrogate, display) — see _TransitionTask."""
        from spacy_ray_amd.utils import timing

        t2v = task.t2v
        if not task.train:
            return None, 0.0
        if task.cpp or task.gpu is not None:
            outs = [o for o in task.cpp_outs if o and o[0].shape[0] > 0]
            if not outs:
                return t2v.new_zeros(()), 0.0
            if len(outs) == 1:
                scores, gold, valid, feats, which, hidden = outs[0]
            else:
                scores, gold, valid, feats, which, hidden = (
                    torch.cat(c, dim=0) for c in zip(*outs)
                )
            from spacy_ray_amd.ops.api import transition_loop_loss

            mod = self.module
            lkw = {}
            if task.gpu is not None:  # doc-major arenas: atomic-free scatter
                lkw = dict(doc_off=task.gpu["off"], doc_lens=task.gpu["lens"],
                           cap_mult=task.gpu["cap_mult"],
                           maxlen=task.gpu["maxlen"],
                           doc_total=task.gpu["total"])
            with timing.span("raw/loss_build"):
                loss = transition_loop_loss(
                    task.pre, mod.lower_b, mod.upper.weight, mod.upper.bias,
                    scores, gold, valid, feats, which, hidden, **lkw,
                )
                if task.steps_dev is not None:
                    # GPU state machine: arenas are CAPACITY-sized (unused
                    # rows masked out of the CE); normalize by the REAL
                    # transition count the kernel accumulated — a device
                    # scalar, so no host sync
                    loss = loss / task.steps_dev.to(loss.dtype).clamp(min=1)[0]
                else:
                    loss = loss / scores.shape[0]
            return loss, loss.detach()
        if not task.score_chunks:
            return t2v.new_zeros(()), 0.0
        with timing.span("raw/loss_build"):
            all_scores = torch.cat(task.score_chunks, dim=0).float()
            all_gold = torch.cat(task.gold_chunks, dim=0) > 0
            all_valid = torch.cat(task.valid_chunks, dim=0) > 0
            counts_t = all_gold.sum(dim=-1)
            ok_t = counts_t > 0
            logp = torch.log_softmax(
                all_scores.masked_fill(~all_valid, NEG_INF), dim=-1
            )
            target = all_gold.float() / counts_t.clamp(min=1).unsqueeze(-1)
            row_loss = -(target * logp).sum(dim=-1)
            step_loss = row_loss.masked_fill(~ok_t, 0).sum() / max(1, task.n_states_total)
        display = step_loss.detach()
        with timing.span("raw/phase1_bwd"):
            step_loss.backward()  # phase 1: upper + lower_b grads
            if task.fused:
                # the C++ autograd node stashed (feats, dSummed) per step in a
                # GIL-free session store; drain into the batched scatter path
                task.entries.extend(
                    (e[0], e[1]) for e in task.hip.fused_entries_take(task.task_id)
                )
            if task.entries is not None:
                # batched dPre scatter: one sort + segmented reduction for
                # ALL steps (replaces per-step atomic scatters)
                from spacy_ray_amd.ops.api import parser_scatter_entries

                parser_scatter_entries(task.entries, task.dPre32)
                task.entries.clear()
        # dPre32 already carries the 1/n_states normalization (it was filled
        # by the normalized step_loss backward).  inject_grad hands it to the
        # precompute tensor without materializing a surrogate product.
        from spacy_ray_amd.ops.api import inject_grad

        surrogate = inject_grad(task.pre, task.dPre32)
        return surrogate, display

    def _step_loop(self, shards, t2v, train: bool):
        task = self.begin_task(shards, t2v, train)
        run_transition_tasks([task])
        return self.finish_task(task)

    def stage_gold(self, examples, batch) -> None:
        """Concatenate the per-doc cached gold arrays ONCE per batch and cache
        on the TokenBatch: np.concatenate over 50k tiny arrays cost
        ~80 ms/step at 1M words when rebuilt every step (replayed batches
        keep the cache across steps)."""
        if batch is None:
            return
        key = (self.name + "_gold", tuple(self.labels))
        if key not in batch.staged:
            batch.staged[key] = self._build_gold(examples)

    def _build_gold(self, examples):
        raise NotImplementedError

    # ---- GPU-resident state machine (srx_gpustate.hip): the ENTIRE greedy
    # transition loop for a batch runs in ONE kernel launch (one wave64 per
    # doc — state, dynamic oracle, scorer and argmax all in LDS/registers),
    # replacing the per-step host round trips of the C++ loop.  Train
    # returns the same arenas the host loop produces, so the batched CE
    # backward is shared; decode returns heads/labels (or BILUO tags)
    # directly.  SRX_GPU_STATES=0 reverts to the host loop.
    def _gpu_ready(self, lengths, t2v) -> bool:
        if not t2v.is_cuda or os.environ.get("SRX_GPU_STATES", "1") != "1":
            return False
        from spacy_ray_amd.ops import api as _ops

        hip = _ops.hip_ext()
        if hip is None or not hasattr(hip, "gpu_arceager"):
            return False
        if self.hidden_width > 64:
            return False
        return self._gpu_supported(lengths, hip)

    def _gpu_supported(self, lengths, hip) -> bool:
        return False

    def _gold_to_device(self, staged, lengths, device):
        raise NotImplementedError

    def _gpu_launch(self, task) -> None:
        raise NotImplementedError

    def _maybe_gpu_task(self, staged, lengths, t2v, batch, train):
        """Build a GPU-state-machine task if supported, else None."""
        if not self._gpu_ready(lengths, t2v):
            return None
        if train:
            dev_key = (self.name + "_gold_dev", tuple(self.labels))
            gdev = batch.staged.get(dev_key) if batch is not None else None
            if gdev is None:
                gdev = self._gold_to_device(staged, lengths, t2v.device)
                if batch is not None:
                    batch.staged[dev_key] = gdev
        else:
            gdev = self._gold_to_device(None, lengths, t2v.device)
        task = self.begin_task([], t2v, train=train)
        task.gpu = gdev
        return task

    def make_loss_task(self, examples, t2v, batch=None) -> "_TransitionTask":
        key = (self.name + "_gold", tuple(self.labels))
        staged = batch.staged.get(key) if batch is not None else None
        if staged is None:
            staged = self._build_gold(examples)
        if batch is not None and len(batch.docs) == len(examples):
            lengths = batch.lengths_np
        else:
            lengths = np.asarray([len(eg.reference) for eg in examples], dtype=np.int32)
        task = self._maybe_gpu_task(staged, lengths, t2v, batch, train=True)
        if task is not None:
            return task
        shards = []
        for lo, hi, base in self._split_docs(lengths, self._n_shards(t2v)):
            states = self._make_states(lengths[lo:hi], base)
            self._set_gold(states, staged)  # global flat gold; offsets select
            shards.append(states)
        return self.begin_task(shards, t2v, train=True)

    def make_predict_task(self, docs, t2v):
        lengths = np.asarray([len(d) for d in docs], dtype=np.int32)
        task = self._maybe_gpu_task(None, lengths, t2v, None, train=False)
        if task is not None:
            return task, None, None
        splits = self._split_docs(lengths, self._n_shards(t2v))
        shards = [self._make_states(lengths[lo:hi], base) for lo, hi, base in splits]
        return self.begin_task(shards, t2v, train=False), splits, shards

    def _n_shards(self, t2v) -> int:
        # Measured on MI355X: 2-way shard pipelining LOST ~25% words/s on
        # 20-word-doc batches — each shard doubles the per-step python/launch
        # overhead, which exceeds the hidden sync latency.  Default 1; the
        # machinery stays for long-doc workloads (SRX_PARSER_SHARDS to tune).
        import os

        return int(os.environ.get("SRX_PARSER_SHARDS", "1")) if t2v.is_cuda else 1

    def get_loss(self, examples, t2v, batch):
        task = self.make_loss_task(examples, t2v, batch)
        run_transition_tasks([task])
        return self.finish_task(task)

    def predict_and_set(self, docs, t2v, batch) -> None:
        if getattr(self, "beam_width", 1) > 1:
            self._beam_annotate(docs, t2v)
            return
        task, splits, shards = self.make_predict_task(docs, t2v)
        with torch.no_grad():
            run_transition_tasks([task])
        if task.gpu is not None:
            self._annotate_gpu(docs, task.gpu_decode)
            return
        for (lo, hi, base), states in zip(splits, shards):
            self._annotate(docs[lo:hi], states)

    def _beam_annotate(self, docs, t2v) -> None:
        raise NotImplementedError

    def _annotate_gpu(self, docs, decode) -> None:
        raise NotImplementedError


class ParserPipe(_TransitionPipeBase):
    name = "parser"

    @staticmethod
    def labels_from(ref):
        return (d for d in ref.deps if d != "ROOT") if ref.deps else ()

    def __init__(self, name: str, spec, use_break: bool = False,
                 beam_width: int = 1) -> None:
        super().__init__(name, spec)
        # spaCy USE_BREAK contract (sentence boundaries learned as a BREAK
        # transition) — see transitions.cpp / docs/PARITY.md; off by default
        self.use_break = use_break
        # decode-time beam search (spaCy's beam parser role; training is
        # greedy like the reference) — pipeline/beam.py
        self.beam_width = int(beam_width)
        if self.beam_width > 1 and use_break:
            raise ValueError("beam decoding does not support use_break")

    def _beam_annotate(self, docs, t2v) -> None:
        from .beam import beam_parse_annotate

        beam_parse_annotate(self, docs, t2v, self.beam_width)

    def initialize(self, examples, device) -> None:
        if not self.labels:
            labels = set()
            for eg in examples:
                if eg.reference.deps:
                    labels.update(d for d in eg.reference.deps if d != "ROOT")
            self.labels = sorted(labels) or ["dep"]
            self.label2id = {t: i for i, t in enumerate(self.labels)}
        self.cfg["labels"] = self.labels
        self.cfg["use_break"] = self.use_break
        self._build_module(device)

    def load_cfg(self, cfg, device) -> None:
        self.use_break = bool(cfg.get("use_break", self.use_break))
        super().load_cfg(cfg, device)

    def _n_actions(self) -> int:
        return 2 + 2 * len(self.labels) + (1 if self.use_break else 0)

    def _make_states(self, lengths, base: int = 0):
        return _srx_cpu.ArcEagerBatch(lengths, len(self.labels), base,
                                      self.use_break)

    def _gold_arrays(self, eg) -> Tuple[np.ndarray, np.ndarray]:
        """Per-example (heads, label-ids), cached on the reference Doc —
        corpora/bench replay docs across steps, and the per-token python
        loop over 1M words cost ~100 ms/step before caching (same trick as
        the tagger's _gold_ids)."""
        ref = eg.reference
        key = ("dep_gold", tuple(self.labels))
        cached = ref.user_data.get(key)
        if cached is None:
            n = len(ref)
            heads = (ref.heads.astype(np.int32) if ref.heads is not None
                     else np.full(n, -1, dtype=np.int32))
            deps = ref.deps or ["dep"] * n
            labs = np.zeros(n, dtype=np.int32)
            l2i = self.label2id
            for i, d in enumerate(deps):
                lid = l2i.get(d)
                if lid is None:
                    if d == "ROOT":
                        lid = 0  # root label never drives a labeled arc cost
                    else:
                        raise ValueError(
                            f"parser gold dep label {d!r} is not in the "
                            f"component's label set (size "
                            f"{len(self.labels)}) — labels are discovered "
                            f"over the full training corpus at init (or "
                            f"pinned via the component's `labels` config)"
                        )
                labs[i] = lid
            cached = (heads, labs)
            ref.user_data[key] = cached
        return cached

    def _build_gold(self, examples):
        pairs = [self._gold_arrays(eg) for eg in examples]
        heads = np.concatenate([p[0] for p in pairs])
        labs = np.concatenate([p[1] for p in pairs])
        sents = None
        if self.use_break and any(eg.reference.sent_starts is not None
                                  for eg in examples):
            sents = np.concatenate([
                eg.reference.sent_starts if eg.reference.sent_starts is not None
                else np.zeros(len(eg.reference), dtype=np.int32)
                for eg in examples
            ]).astype(np.int32)
        return (heads, labs, sents)

    def _set_gold(self, states, staged) -> None:
        states.set_gold(staged[0], staged[1])
        if len(staged) > 2 and staged[2] is not None:
            states.set_sent_gold(staged[2])

    def _annotate(self, docs, states) -> None:
        sents = states.sent_starts() if self.use_break else None
        self._annotate_arrays(docs, states.heads(), states.labels(), sents)

    def _annotate_arrays(self, docs, heads, labels, sents=None) -> None:
        # vectorized label-string lookup over the WHOLE batch (the
        # per-token python loop dominated decode latency at 200k words)
        L = len(self.labels)
        table = np.asarray(list(self.labels) + ["ROOT"], dtype=object)
        labels = np.asarray(labels)
        safe = np.where((labels >= 0) & (labels < L), labels, L)
        dep_strs = np.where(np.asarray(heads) == -1, "ROOT",
                            table[safe]).tolist()  # ONE tolist; per-doc
        # slicing below is C-level list copying (10k numpy slice+tolist
        # calls measured slower than one conversion at serve batch sizes)
        off = 0
        for doc in docs:
            n = len(doc)
            doc.heads = heads[off:off + n].copy()
            doc.deps = dep_strs[off:off + n]
            if sents is not None:
                ss = sents[off:off + n].copy()
                if n > 0:
                    ss[0] = 1  # the first token always starts a sentence
                doc.sent_starts = ss
            off += n

    # ---- GPU state machine hooks (srx_gpustate.hip::gpu_arceager_kernel)
    def _gpu_supported(self, lengths, hip) -> bool:
        # BREAK needs the sentence-gold machinery (host loop); stack/arc
        # arrays live in per-wave LDS sized for GPU_STATE_MAXLEN tokens
        return (not self.use_break and
                (len(lengths) == 0 or int(lengths.max()) <= hip.GPU_STATE_MAXLEN))

    def _gold_to_device(self, staged, lengths, device):
        lengths = np.asarray(lengths, dtype=np.int64)
        n_docs = len(lengths)
        off = np.zeros(n_docs, dtype=np.int64)
        if n_docs > 1:
            np.cumsum(lengths[:-1], out=off[1:])
        total = int(lengths.sum())

        def to(a, dt):
            return torch.from_numpy(np.ascontiguousarray(a, dtype=dt)).to(device)

        g = {
            "off": to(off, np.int32), "lens": to(lengths, np.int32),
            "total": total, "cap_mult": 2,
            "maxlen": int(lengths.max()) if n_docs else 0,
        }
        empty = torch.empty(0, dtype=torch.int32, device=device)
        if staged is None:  # decode: the oracle inputs are never read
            g.update(gh=empty, gl=empty, kids_off=empty, kids=empty)
            return g
        heads, labs = staged[0], staged[1]
        # gold-children CSR over batch-global parent ids with DOC-LOCAL
        # child ids (mirrors ArcEagerBatch::set_gold's kids/kids_off)
        tok_off = np.repeat(off, lengths)
        tok_len = np.repeat(lengths, lengths)
        local = np.arange(total, dtype=np.int64) - tok_off
        gh = heads.astype(np.int64)
        ok = (gh >= 0) & (gh < tok_len)
        parent = tok_off[ok] + gh[ok]
        order = np.argsort(parent, kind="stable")
        kids = local[ok][order]
        kids_off = np.zeros(total + 1, dtype=np.int64)
        np.cumsum(np.bincount(parent, minlength=total), out=kids_off[1:])
        g.update(gh=to(heads, np.int32), gl=to(labs, np.int32),
                 kids_off=to(kids_off, np.int32), kids=to(kids, np.int32))
        return g

    def _gpu_launch(self, task) -> None:
        g = task.gpu
        mod = self.module
        outs = task.hip.gpu_arceager(
            task.pre_d.contiguous(), g["off"], g["lens"], g["gh"], g["gl"],
            g["kids_off"], g["kids"], mod.lower_b.detach(),
            mod.upper.weight.detach(), mod.upper.bias.detach(), g["total"],
            len(self.labels), task.train,
        )
        if task.train:
            task.cpp_outs.append(list(outs[:6]))
            task.steps_dev = outs[8]
        else:
            task.gpu_decode = (outs[6], outs[7])

    def _annotate_gpu(self, docs, decode) -> None:
        heads_d, labels_d = decode
        self._annotate_arrays(
            docs, heads_d.cpu().numpy(), labels_d.cpu().numpy()
        )


class NerPipe(_TransitionPipeBase):
    name = "ner"

    @staticmethod
    def labels_from(ref):
        if not ref.ents:
            return ()
        return (t.partition("-")[2] for t in ref.ents
                if t not in ("O", "-", None, ""))

    def __init__(self, name: str, spec, beam_width: int = 1) -> None:
        super().__init__(name, spec)
        self.beam_width = int(beam_width)

    def _beam_annotate(self, docs, t2v) -> None:
        from .beam import beam_ner_annotate

        beam_ner_annotate(self, docs, t2v, self.beam_width)

    def initialize(self, examples, device) -> None:
        if not self.labels:
            labels = set()
            for eg in examples:
                if eg.reference.ents:
                    for tag in eg.reference.ents:
                        if tag not in ("O", "-", None, ""):
                            labels.add(tag.partition("-")[2])
            self.labels = sorted(labels) or ["ENT"]
            self.label2id = {t: i for i, t in enumerate(self.labels)}
        self.cfg["labels"] = self.labels
        self._build_module(device)

    def _n_actions(self) -> int:
        return 1 + 4 * len(self.labels)

    def _make_states(self, lengths, base: int = 0):
        return _srx_cpu.BiluoBatch(lengths, len(self.labels), base)

    def _gold_codes(self, eg) -> np.ndarray:
        # cached per reference Doc (see ParserPipe._gold_arrays)
        ref = eg.reference
        key = ("biluo_gold", tuple(self.labels))
        cached = ref.user_data.get(key)
        if cached is None:
            cached = biluo_to_codes(
                ref.ents or ["O"] * len(ref), self.label2id
            ).astype(np.int32)
            ref.user_data[key] = cached
        return cached

    def _build_gold(self, examples):
        return np.concatenate([self._gold_codes(eg) for eg in examples])

    def _set_gold(self, states, staged) -> None:
        states.set_gold(staged)

    def _annotate(self, docs, states) -> None:
        self._annotate_tags(docs, states.tags())

    def _annotate_tags(self, docs, tags) -> None:
        # one vectorized code->string lookup for the whole batch
        from spacy_ray_amd.vocab.doc import biluo_string_table

        table = biluo_string_table(self.labels)
        tags = np.asarray(tags)
        safe = np.where((tags > 0) & (tags < len(table)), tags, 0)
        ent_strs = table[safe].tolist()
        off = 0
        for doc in docs:
            n = len(doc)
            doc.ents = ent_strs[off:off + n]
            off += n

    # ---- GPU state machine hooks (srx_gpustate.hip::gpu_biluo_kernel)
    def _gpu_supported(self, lengths, hip) -> bool:
        return True  # (open, open_start) state is register-resident: no len cap

    def _gold_to_device(self, staged, lengths, device):
        lengths = np.asarray(lengths, dtype=np.int64)
        n_docs = len(lengths)
        off = np.zeros(n_docs, dtype=np.int64)
        if n_docs > 1:
            np.cumsum(lengths[:-1], out=off[1:])

        def to(a, dt):
            return torch.from_numpy(np.ascontiguousarray(a, dtype=dt)).to(device)

        gold = (to(staged, np.int32) if staged is not None
                else torch.empty(0, dtype=torch.int32, device=device))
        return {"off": to(off, np.int32), "lens": to(lengths, np.int32),
                "total": int(lengths.sum()), "gold": gold, "cap_mult": 1,
                "maxlen": int(lengths.max()) if n_docs else 0}

    def _gpu_launch(self, task) -> None:
        g = task.gpu
        mod = self.module
        outs = task.hip.gpu_biluo(
            task.pre_d.contiguous(), g["off"], g["lens"], g["gold"],
            mod.lower_b.detach(), mod.upper.weight.detach(),
            mod.upper.bias.detach(), g["total"], len(self.labels), task.train,
        )
        if task.train:
            task.cpp_outs.append(list(outs[:6]))
            task.steps_dev = outs[7]
        else:
            task.gpu_decode = outs[6]

    def _annotate_gpu(self, docs, decode) -> None:
        self._annotate_tags(docs, decode.cpu().numpy())


class TextcatHead(nn.Module):
    def __init__(self, width: int, n_cats: int):
        super().__init__()
        self.output = nn.Linear(width, n_cats)
        nn.init.zeros_(self.output.weight)
        nn.init.zeros_(self.output.bias)

    def forward(self, X):
        from spacy_ray_amd.ops.api import linear_cdw

        return linear_cdw(X, self.output.weight, self.output.bias)


class TextcatPipe(TrainablePipe):
    """Doc-level classification over mean-pooled tok2vec (the reference is
    a generic spaCy trainer — `spacy ray train` accepts textcat configs,
    /root/reference/worker.py:71-106 builds whatever pipeline the config
    names).  `exclusive_classes=True` = spaCy's `textcat` (softmax CE vs
    the gold distribution); False = `textcat_multilabel` (per-label
    binary cross-entropy).  Pooling = reduce_mean_ragged (the segmented
    reduce kernel)."""

    name = "textcat"
    listens_to = "tok2vec"

    @staticmethod
    def labels_from(ref):
        return ref.cats.keys() if ref.cats else ()

    def __init__(self, name: str, spec, exclusive_classes: bool = True) -> None:
        super().__init__()
        self.name = name
        self.width = spec.width
        self.embedded_spec = getattr(spec, "embedded_tok2vec", None)
        self.exclusive = exclusive_classes
        self.labels: List[str] = []
        self.label2id: Dict[str, int] = {}

    def initialize(self, examples, device) -> None:
        if not self.labels:
            labels = set()
            for eg in examples:
                if eg.reference.cats:
                    labels.update(eg.reference.cats.keys())
            self.labels = sorted(labels) or ["POSITIVE"]
            self.label2id = {t: i for i, t in enumerate(self.labels)}
        self.cfg["labels"] = self.labels
        self.cfg["exclusive"] = self.exclusive
        if self.module is None:
            self.module = TextcatHead(self.width, max(1, len(self.labels))).to(device)
        self._attach_embedded(device)

    def load_cfg(self, cfg, device) -> None:
        super().load_cfg(cfg, device)
        self.labels = list(cfg.get("labels", []))
        self.label2id = {t: i for i, t in enumerate(self.labels)}
        self.exclusive = bool(cfg.get("exclusive", True))
        if self.module is None:
            self.module = TextcatHead(self.width, max(1, len(self.labels))).to(device)
        self._attach_embedded(device)

    def _doc_scores(self, t2v, batch, docs):
        from spacy_ray_amd.ops import api as _ops

        if batch is not None:
            lengths, total = batch.lengths, batch.n_tokens
        else:
            lens = [len(d) for d in docs]
            lengths = torch.as_tensor(lens, dtype=torch.int64,
                                      device=t2v.device)
            total = sum(lens)
        pooled = _ops.reduce_mean_ragged(t2v[:total], lengths)
        # TokenBatch may append a PAD pseudo-doc (GEMM shape bucketing)
        return self.module(pooled[:len(docs)])

    def _gold_matrix(self, examples) -> np.ndarray:
        n = len(examples)
        Y = np.zeros((n, max(1, len(self.labels))), dtype=np.float32)
        for i, eg in enumerate(examples):
            for lab, val in (eg.reference.cats or {}).items():
                j = self.label2id.get(lab)
                if j is not None:
                    Y[i, j] = float(val)
        return Y

    def get_loss(self, examples, t2v, batch):
        scores = self._doc_scores(t2v, batch, [eg.reference for eg in examples])
        gold = torch.from_numpy(self._gold_matrix(examples)).to(scores.device)
        n = max(1, scores.shape[0])
        if self.exclusive:
            logp = torch.log_softmax(scores.float(), dim=-1)
            loss = -(gold * logp).sum() / n
        else:
            loss = torch.nn.functional.binary_cross_entropy_with_logits(
                scores.float(), gold, reduction="sum") / n
        return loss, loss.detach()

    def predict_and_set(self, docs, t2v, batch) -> None:
        with torch.no_grad():
            scores = self._doc_scores(t2v, batch, docs)
            probs = (torch.softmax(scores.float(), dim=-1) if self.exclusive
                     else torch.sigmoid(scores.float())).cpu().numpy()
        for i, doc in enumerate(docs):
            doc.cats = {lab: float(probs[i, j])
                        for j, lab in enumerate(self.labels)}

    def state_cfg(self) -> Dict:
        return self.cfg


class SenterPipe(TaggerPipe):
    """Sentence-boundary recognizer (spaCy's `senter`): a 2-class per-token
    head over sent_starts — reuses the tagger machinery with gold derived
    from `Doc.sent_starts` and predictions written back there."""

    name = "senter"

    @staticmethod
    def labels_from(ref):
        return ("I", "S")  # fixed label set

    def initialize(self, examples, device) -> None:
        self.labels = ["I", "S"]  # S = sentence start
        self.label2id = {"I": 0, "S": 1}
        self.cfg["labels"] = self.labels
        if self.module is None:
            self.module = TaggerHead(self.width, 2).to(device)
        self._attach_embedded(device)

    def _gold_ids(self, examples, n_tokens: int) -> np.ndarray:
        key = ("senter_ids", self.name)
        parts = []
        for eg in examples:
            ref = eg.reference
            cached = ref.user_data.get(key)
            if cached is None:
                n = len(ref)
                if ref.sent_starts is not None:
                    cached = (ref.sent_starts > 0).astype(np.int64)
                    if n > 0:
                        cached = cached.copy()
                        cached[0] = 1  # the first token always starts one
                else:
                    cached = np.full(n, -1, dtype=np.int64)  # unannotated
                ref.user_data[key] = cached
            parts.append(cached)
        ids = np.concatenate(parts) if parts else np.zeros(0, dtype=np.int64)
        if len(ids) < n_tokens:
            ids = np.concatenate([ids, np.full(n_tokens - len(ids), -1,
                                               dtype=np.int64)])
        return ids

    def stage_gold(self, examples, batch) -> None:
        key = ("tagger_gold", self.name)
        if batch is None or key in batch.staged:
            return
        from spacy_ray_amd.utils.pinned import to_device

        gold_np = self._gold_ids(examples, batch.n_tokens)
        gold = to_device(gold_np, batch.attr_ids.device)
        batch.staged[key] = (gold, int((gold_np >= 0).sum()))

    def predict_and_set(self, docs, t2v, batch) -> None:
        with torch.no_grad():
            pred = self.module(t2v).argmax(dim=-1).cpu().numpy()
        off = 0
        for doc in docs:
            n = len(doc)
            ss = pred[off:off + n].astype(np.int32)
            if n > 0:
                ss[0] = 1
            doc.sent_starts = ss
            off += n


class MorphologizerPipe(TaggerPipe):
    """Morphological feature prediction (spaCy's `morphologizer`): a
    per-token classifier over the observed UD FEATS strings (Doc.morphs;
    populated by `spacy-mi convert` from CoNLL-U column 6).  Reuses the
    tagger head/loss machinery; predictions land back in Doc.morphs and
    score as morph_acc."""

    name = "morphologizer"

    @staticmethod
    def labels_from(ref):
        return (m for m in ref.morphs if m) if ref.morphs else ()

    def initialize(self, examples, device) -> None:
        if not self.labels:
            labels = set()
            for eg in examples:
                if eg.reference.morphs:
                    labels.update(m for m in eg.reference.morphs if m)
            self.labels = sorted(labels) or ["_"]
            self.label2id = {t: i for i, t in enumerate(self.labels)}
        self.cfg["labels"] = self.labels
        if self.module is None:
            self.module = TaggerHead(self.width, max(1, len(self.labels))).to(device)
        self._attach_embedded(device)

    def _gold_ids(self, examples, n_tokens: int) -> np.ndarray:
        key = ("morph_ids", tuple(self.labels))
        parts = []
        for eg in examples:
            ref = eg.reference
            cached = ref.user_data.get(key)
            if cached is None:
                morphs = ref.morphs or ["" for _ in range(len(ref))]
                cached = np.fromiter(
                    (self.label2id.get(m, -1) for m in morphs),
                    dtype=np.int64, count=len(morphs),
                )
                ref.user_data[key] = cached
            parts.append(cached)
        ids = np.concatenate(parts) if parts else np.zeros(0, dtype=np.int64)
        if len(ids) < n_tokens:
            ids = np.concatenate([ids, np.full(n_tokens - len(ids), -1,
                                               dtype=np.int64)])
        return ids

    def stage_gold(self, examples, batch) -> None:
        key = ("tagger_gold", self.name)
        if batch is None or key in batch.staged:
            return
        from spacy_ray_amd.utils.pinned import to_device

        gold_np = self._gold_ids(examples, batch.n_tokens)
        gold = to_device(gold_np, batch.attr_ids.device)
        batch.staged[key] = (gold, int((gold_np >= 0).sum()))

    def predict_and_set(self, docs, t2v, batch) -> None:
        with torch.no_grad():
            pred = self.module(t2v).argmax(dim=-1).cpu().numpy()
        table = (np.asarray(self.labels, dtype=object) if self.labels
                 else np.asarray([""], dtype=object))
        morph_strs = table[np.clip(pred, 0, len(table) - 1)]
        off = 0
        for doc in docs:
            n = len(doc)
            doc.morphs = morph_strs[off:off + n].tolist()
            off += n


class SpancatHead(nn.Module):
    def __init__(self, width: int, n_labels: int):
        super().__init__()
        self.output = nn.Linear(width, n_labels)
        nn.init.zeros_(self.output.weight)
        nn.init.zeros_(self.output.bias)

    def forward(self, X):
        from spacy_ray_amd.ops.api import linear_cdw

        return linear_cdw(X, self.output.weight, self.output.bias)


class SpancatPipe(TrainablePipe):
    """Span categorizer (spaCy's `spancat`): an ngram suggester (sizes
    1..max_ngram) over each doc plus a multi-label classifier on pooled
    span representations.  Overlapping spans allowed (the point of spancat
    vs NER).  Span pooling is mean-over-tokens computed from ONE prefix
    sum of the tok2vec matrix — every candidate is two gathers and a
    divide, no per-span loops.  Gold/predictions live in
    `Doc.spans[spans_key]` as (start, end, label) triples; scored as
    spans_sc_p/r/f."""

    name = "spancat"
    listens_to = "tok2vec"

    def labels_from(self, ref):
        return (lab for (_s, _e, lab) in ref.spans.get(self.spans_key, ()))

    def __init__(self, name: str, spec, spans_key: str = "sc",
                 max_ngram: int = 3, threshold: float = 0.5) -> None:
        super().__init__()
        self.name = name
        self.width = spec.width
        self.embedded_spec = getattr(spec, "embedded_tok2vec", None)
        self.spans_key = spans_key
        self.max_ngram = int(max_ngram)
        self.threshold = float(threshold)
        self.labels: List[str] = []
        self.label2id: Dict[str, int] = {}

    def initialize(self, examples, device) -> None:
        if not self.labels:
            labels = set()
            for eg in examples:
                for (s, e, lab) in eg.reference.spans.get(self.spans_key, []):
                    labels.add(lab)
            self.labels = sorted(labels) or ["SPAN"]
            self.label2id = {t: i for i, t in enumerate(self.labels)}
        self.cfg.update(labels=self.labels, spans_key=self.spans_key,
                        max_ngram=self.max_ngram, threshold=self.threshold)
        if self.module is None:
            self.module = SpancatHead(self.width, max(1, len(self.labels))).to(device)
        self._attach_embedded(device)

    def load_cfg(self, cfg, device) -> None:
        super().load_cfg(cfg, device)
        self.labels = list(cfg.get("labels", []))
        self.label2id = {t: i for i, t in enumerate(self.labels)}
        self.spans_key = cfg.get("spans_key", "sc")
        self.max_ngram = int(cfg.get("max_ngram", 3))
        self.threshold = float(cfg.get("threshold", 0.5))
        if self.module is None:
            self.module = SpancatHead(self.width, max(1, len(self.labels))).to(device)
        self._attach_embedded(device)

    # ---------------------------------------------------------- candidates
    def _suggest(self, lengths: np.ndarray):
        """ngram candidates -> (doc_idx, start, end) arrays + global token
        offsets; deterministic order (doc-major, then ngram-size blocks,
        start-major) — the gold indexer's closed form depends on it.
        Cached by the length vector (training replays fixed batches)."""
        key = lengths.tobytes()
        cached = getattr(self, "_sugg_cache", None)
        if cached is not None and cached[0] == key:
            return cached[1]
        di, ss, ee = [], [], []
        off = 0
        for d, n in enumerate(lengths.tolist()):
            for k in range(1, self.max_ngram + 1):
                for s in range(0, n - k + 1):
                    di.append(d)
                    ss.append(off + s)
                    ee.append(off + s + k)
            off += n
        out = (np.asarray(di, dtype=np.int64), np.asarray(ss, dtype=np.int64),
               np.asarray(ee, dtype=np.int64))
        self._sugg_cache = (key, out)
        return out

    def _span_scores(self, t2v, lengths: np.ndarray):
        device = t2v.device
        di, ss, ee = self._suggest(lengths)
        total = int(lengths.sum())
        csum = torch.cumsum(t2v[:total].float(), dim=0)
        P = torch.cat([csum.new_zeros(1, csum.shape[1]), csum], dim=0)
        s_t = torch.from_numpy(ss).to(device)
        e_t = torch.from_numpy(ee).to(device)
        pooled = (P[e_t] - P[s_t]) / (e_t - s_t).unsqueeze(1).float()
        return self.module(pooled.to(t2v.dtype)), di, ss, ee

    def get_loss(self, examples, t2v, batch):
        lengths = (batch.lengths_np if batch is not None
                   and len(batch.docs) == len(examples)
                   else np.asarray([len(eg.reference) for eg in examples],
                                   dtype=np.int32))
        scores, di, ss, ee = self._span_scores(t2v, lengths)
        # gold target: O(#gold spans) — a candidate's index is closed-form
        # (doc base + ngram-size block offset + start), so each gold span
        # hits exactly one row instead of scanning all candidates x labels
        Y = np.zeros((len(di), max(1, len(self.labels))), dtype=np.float32)
        n_per_doc = lengths.astype(np.int64)
        cand_per_doc = np.zeros(len(lengths), dtype=np.int64)
        for k in range(1, self.max_ngram + 1):
            cand_per_doc += np.maximum(0, n_per_doc - k + 1)
        doc_base = np.zeros(len(lengths), dtype=np.int64)
        if len(lengths) > 1:
            np.cumsum(cand_per_doc[:-1], out=doc_base[1:])
        for d, eg in enumerate(examples):
            n = int(n_per_doc[d])
            for (s0, e0, lab) in eg.reference.spans.get(self.spans_key, []):
                k = int(e0) - int(s0)
                j = self.label2id.get(lab)
                if j is None or k < 1 or k > self.max_ngram or int(e0) > n:
                    continue
                block = sum(max(0, n - kk + 1) for kk in range(1, k))
                Y[doc_base[d] + block + int(s0), j] = 1.0
        target = torch.from_numpy(Y).to(scores.device)
        n = max(1, scores.shape[0])
        loss = torch.nn.functional.binary_cross_entropy_with_logits(
            scores.float(), target, reduction="sum") / n
        return loss, loss.detach()

    def predict_and_set(self, docs, t2v, batch) -> None:
        lengths = np.asarray([len(d) for d in docs], dtype=np.int32)
        with torch.no_grad():
            scores, di, ss, ee = self._span_scores(t2v, lengths)
            probs = torch.sigmoid(scores.float()).cpu().numpy()
        offs = np.zeros(len(lengths), dtype=np.int64)
        np.cumsum(lengths[:-1], out=offs[1:]) if len(lengths) > 1 else None
        found = [[] for _ in docs]
        hit_c, hit_j = np.nonzero(probs > self.threshold)
        for c, j in zip(hit_c.tolist(), hit_j.tolist()):
            d = int(di[c])
            found[d].append((int(ss[c] - offs[d]), int(ee[c] - offs[d]),
                             self.labels[j]))
        for d, doc in enumerate(docs):
            doc.spans[self.spans_key] = found[d]
