"""Beam-search decoding for the transition pipes (spaCy ships beam
variants of the parser/NER; the reference trains greedy only, so this is
a decode-time accuracy option, not a training path).

Design: pure-python transition states (decode-only mirrors of the C++
systems in ops/csrc/transitions.cpp — same features, valid rules and
advance semantics, parity-tested against the greedy path at width 1)
with BATCHED scoring: every live candidate across all docs is scored in
one parser_step_score + upper GEMM per step, so the GPU sees one fused
gather/maxout kernel and one GEMM per beam step instead of per-candidate
work.  Candidates expand by per-action log-softmax-over-valid scores;
each doc keeps its `width` best by total log-probability.

Enable with ``beam_width = N`` on the parser/ner component config
(``factory = "parser"`` block); width 1 short-circuits to the greedy
decoder.
"""
from __future__ import annotations

from typing import List, Tuple

import numpy as np
import torch


class _ArcEagerState:
    """Decode-only arc-eager state (mirror of transitions.cpp semantics:
    fill_features 13 / fill_valid / apply_action, no BREAK)."""

    __slots__ = ("n", "stack", "buf", "head", "label", "l1", "l2", "r1",
                 "r2", "score")

    def __init__(self, n: int):
        self.n = n
        self.stack: List[int] = []
        self.buf = 0
        self.head = [-1] * n
        self.label = [-1] * n
        self.l1 = [-1] * n
        self.l2 = [-1] * n
        self.r1 = [-1] * n
        self.r2 = [-1] * n
        self.score = 0.0

    def clone(self) -> "_ArcEagerState":
        s = _ArcEagerState.__new__(_ArcEagerState)
        s.n = self.n
        s.stack = list(self.stack)
        s.buf = self.buf
        s.head = list(self.head)
        s.label = list(self.label)
        s.l1 = list(self.l1)
        s.l2 = list(self.l2)
        s.r1 = list(self.r1)
        s.r2 = list(self.r2)
        s.score = self.score
        return s

    def is_final(self) -> bool:
        return self.buf >= self.n and len(self.stack) <= 1

    def features(self) -> List[int]:
        st, n, buf = self.stack, self.n, self.buf
        s0 = st[-1] if st else -1
        s1 = st[-2] if len(st) > 1 else -1
        s2 = st[-3] if len(st) > 2 else -1
        f = [s0, s1, s2,
             buf if buf < n else -1,
             buf + 1 if buf + 1 < n else -1,
             buf + 2 if buf + 2 < n else -1,
             self.l1[s0] if s0 >= 0 else -1,
             self.l2[s0] if s0 >= 0 else -1,
             self.r1[s0] if s0 >= 0 else -1,
             self.r2[s0] if s0 >= 0 else -1,
             self.l1[s1] if s1 >= 0 else -1,
             self.r1[s1] if s1 >= 0 else -1,
             self.head[s0] if s0 >= 0 else -1]
        return f

    def valid_mask(self, n_labels: int) -> np.ndarray:
        A = 2 + 2 * n_labels
        v = np.zeros(A, dtype=bool)
        has_buf = self.buf < self.n
        has_s0 = bool(self.stack)
        s0_has_head = has_s0 and self.head[self.stack[-1]] != -1
        if has_buf:
            v[0] = True
        if has_s0 and (s0_has_head or not has_buf):
            v[1] = True
        if has_s0 and has_buf and not s0_has_head:
            v[2:2 + n_labels] = True
        if has_s0 and has_buf:
            v[2 + n_labels:2 + 2 * n_labels] = True
        return v

    def apply(self, act: int, n_labels: int) -> None:
        if act == 0:  # SHIFT
            self.stack.append(self.buf)
            self.buf += 1
        elif act == 1:  # REDUCE
            self.stack.pop()
        elif act < 2 + n_labels:  # LEFT-ARC
            dep = self.stack.pop()
            h = self.buf
            self.head[dep] = h
            self.label[dep] = act - 2
            if self.l1[h] == -1 or dep < self.l1[h]:
                self.l2[h] = self.l1[h]
                self.l1[h] = dep
            elif self.l2[h] == -1 or dep < self.l2[h]:
                self.l2[h] = dep
        else:  # RIGHT-ARC
            h = self.stack[-1]
            dep = self.buf
            self.head[dep] = h
            self.label[dep] = act - 2 - n_labels
            if self.r1[h] == -1 or dep > self.r1[h]:
                self.r2[h] = self.r1[h]
                self.r1[h] = dep
            elif self.r2[h] == -1 or dep > self.r2[h]:
                self.r2[h] = dep
            self.stack.append(dep)
            self.buf += 1


class _BiluoState:
    """Decode-only BILUO state (mirror of transitions.cpp BiluoBatch)."""

    __slots__ = ("n", "i", "open", "open_start", "tags", "score")

    def __init__(self, n: int):
        self.n = n
        self.i = 0
        self.open = -1
        self.open_start = -1
        self.tags = [0] * n
        self.score = 0.0

    def clone(self) -> "_BiluoState":
        s = _BiluoState.__new__(_BiluoState)
        s.n = self.n
        s.i = self.i
        s.open = self.open
        s.open_start = self.open_start
        s.tags = list(self.tags)
        s.score = self.score
        return s

    def is_final(self) -> bool:
        return self.i >= self.n

    def features(self) -> List[int]:
        i, n = self.i, self.n
        f = [i - 2, i - 1, i, i + 1, i + 2, self.open_start]
        return [x if 0 <= x < n else -1 for x in f]

    def valid_mask(self, n_types: int) -> np.ndarray:
        A = 1 + 4 * n_types
        v = np.zeros(A, dtype=bool)
        last = self.i == self.n - 1
        if self.open < 0:
            v[0] = True
            for t in range(n_types):
                if not last:
                    v[1 + 4 * t] = True  # B
                v[4 + 4 * t] = True      # U
        else:
            if not last:
                v[2 + 4 * self.open] = True  # I
            v[3 + 4 * self.open] = True      # L
        return v

    def apply(self, act: int, n_types: int) -> None:
        self.tags[self.i] = act
        if act == 0:
            self.open = -1
            self.open_start = -1
        else:
            kind = (act - 1) % 4
            t = (act - 1) // 4
            if kind == 0:
                self.open = t
                self.open_start = self.i
            elif kind != 1:
                self.open = -1
                self.open_start = -1
        self.i += 1


def beam_decode(pipe, docs, t2v, width: int, state_cls):
    """Generic beam driver.  Returns the best final state per doc."""
    from spacy_ray_amd.ops import api as _ops

    mod = pipe.module
    n_labels = len(pipe.labels)
    nF = 13 if state_cls is _ArcEagerState else 6
    with torch.no_grad():
        pre = mod.precompute(t2v).detach().contiguous()
    T1 = pre.shape[0]
    pad_row = T1 - 1
    lengths = [len(d) for d in docs]
    offs = np.concatenate([[0], np.cumsum(lengths[:-1])]).astype(np.int64) \
        if len(lengths) > 1 else np.zeros(1, dtype=np.int64)
    beams: List[List] = [[state_cls(n)] for n in lengths]
    device = t2v.device
    while True:
        live: List[Tuple[int, int]] = [
            (d, k) for d, beam in enumerate(beams)
            for k, st in enumerate(beam) if not st.is_final()
        ]
        if not live:
            break
        feats = np.empty((len(live), nF), dtype=np.int64)
        for row, (d, k) in enumerate(live):
            f = beams[d][k].features()
            feats[row] = [offs[d] + x if x >= 0 else pad_row for x in f]
        feats_t = torch.from_numpy(feats).to(device)
        with torch.no_grad():
            hidden = _ops.parser_step_score(pre, feats_t, mod.lower_b)
            scores = mod.upper(hidden).float().cpu().numpy()
        # per-doc expansion
        by_doc: dict = {}
        for row, (d, k) in enumerate(live):
            by_doc.setdefault(d, []).append((k, scores[row]))
        for d, rows in by_doc.items():
            cands = []
            # finished candidates carry over unchanged
            for k, st in enumerate(beams[d]):
                if st.is_final():
                    cands.append((st.score, None, st, -1))
            for k, srow in rows:
                st = beams[d][k]
                v = st.valid_mask(n_labels)
                masked = np.where(v, srow, -1e30)
                m = masked.max()
                logp = masked - (m + np.log(np.exp(masked - m).sum()))
                for a in np.flatnonzero(v):
                    cands.append((st.score + float(logp[a]), k, st, int(a)))
            cands.sort(key=lambda c: -c[0])
            new_beam = []
            for sc, k, st, a in cands[:width]:
                if a < 0:
                    new_beam.append(st)
                else:
                    ns = st.clone()
                    ns.apply(a, n_labels)
                    ns.score = sc
                    new_beam.append(ns)
            beams[d] = new_beam
    return [max(beam, key=lambda s: s.score) for beam in beams]


def beam_parse_annotate(pipe, docs, t2v, width: int) -> None:
    best = beam_decode(pipe, docs, t2v, width, _ArcEagerState)
    heads = np.concatenate([np.asarray(s.head, dtype=np.int32) for s in best]) \
        if best else np.zeros(0, dtype=np.int32)
    labels = np.concatenate([np.asarray(s.label, dtype=np.int32) for s in best]) \
        if best else np.zeros(0, dtype=np.int32)
    pipe._annotate_arrays(docs, heads, labels)


def beam_ner_annotate(pipe, docs, t2v, width: int) -> None:
    best = beam_decode(pipe, docs, t2v, width, _BiluoState)
    tags = np.concatenate([np.asarray(s.tags, dtype=np.int32) for s in best]) \
        if best else np.zeros(0, dtype=np.int32)
    pipe._annotate_tags(docs, tags)
