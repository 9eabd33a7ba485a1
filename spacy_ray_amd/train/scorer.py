"""Scoring: tag accuracy, UAS/LAS, NER P/R/F (spaCy Scorer contract for the
score keys the reference's loggers/score_weights consume)."""
from __future__ import annotations

from typing import Dict, List, Sequence, Set, Tuple

from spacy_ray_amd.vocab.doc import Example


def _ents_to_spans(ents: List[str]) -> Set[Tuple[int, int, str]]:
    spans = set()
    if not ents:
        return spans
    start, label = None, None
    for i, tag in enumerate(ents):
        if tag is None or tag == "O" or tag == "-" or tag == "":
            start, label = None, None
            continue
        kind, _, lab = tag.partition("-")
        if kind == "U":
            spans.add((i, i + 1, lab))
            start, label = None, None
        elif kind == "B":
            start, label = i, lab
        elif kind == "L" and start is not None and lab == label:
            spans.add((start, i + 1, lab))
            start, label = None, None
        elif kind == "I" and start is not None and lab == label:
            continue
        else:  # inconsistent sequence — drop the open span
            start, label = None, None
    return spans


def score_counts(examples: Sequence[Example], pipe_names: Sequence[str]) -> Dict[str, int]:
    """Additive per-example counts — mergeable across rank shards (the
    distributed eval all-gathers these instead of idling N-1 GPUs while
    rank 0 scores the whole dev set)."""
    c: Dict[str, int] = {}
    if "tagger" in pipe_names:
        correct = total = 0
        for eg in examples:
            gold, pred = eg.reference.tags, eg.predicted.tags
            if gold is None or pred is None:
                continue
            for g, p in zip(gold, pred):
                total += 1
                correct += int(g == p)
        c["tag_correct"], c["tag_total"] = correct, total
    if "parser" in pipe_names:
        uas = las = total = 0
        for eg in examples:
            gh, ph = eg.reference.heads, eg.predicted.heads
            gd, pd = eg.reference.deps, eg.predicted.deps
            if gh is None or ph is None:
                continue
            for i in range(len(eg.reference)):
                # spaCy's Scorer excludes punctuation from dependency
                # metrics (gold dep 'punct'/'p'), so dep_uas/dep_las are
                # comparable to spaCy baselines
                if gd and gd[i] and gd[i].lower() in ("punct", "p"):
                    continue
                total += 1
                if int(gh[i]) == int(ph[i]):
                    uas += 1
                    if gd and pd and gd[i] == pd[i]:
                        las += 1
        c["dep_uas_c"], c["dep_las_c"], c["dep_total"] = uas, las, total
    if "lemmatizer" in pipe_names:
        correct = total = 0
        for eg in examples:
            gold, pred = eg.reference.lemmas, eg.predicted.lemmas
            if gold is None or pred is None:
                continue
            for g, p in zip(gold, pred):
                if not g:
                    continue
                total += 1
                correct += int(g == p)
        c["lemma_correct"], c["lemma_total"] = correct, total
    if "spancat" in pipe_names:
        tp = fp = fn = 0
        for eg in examples:
            for key in (set(eg.reference.spans) | set(eg.predicted.spans)):
                gold = {tuple(sp) for sp in eg.reference.spans.get(key, [])}
                pred = {tuple(sp) for sp in eg.predicted.spans.get(key, [])}
                tp += len(gold & pred)
                fp += len(pred - gold)
                fn += len(gold - pred)
        c["spans_tp"], c["spans_fp"], c["spans_fn"] = tp, fp, fn
    if "morphologizer" in pipe_names:
        correct = total = 0
        for eg in examples:
            gold, pred = eg.reference.morphs, eg.predicted.morphs
            if gold is None or pred is None:
                continue
            for g, p in zip(gold, pred):
                if not g:
                    continue  # missing annotation
                total += 1
                correct += int(g == p)
        c["morph_correct"], c["morph_total"] = correct, total
    if any(n.startswith("textcat") for n in pipe_names):
        correct = total = 0
        for eg in examples:
            gold, pred = eg.reference.cats, eg.predicted.cats
            if not gold or not pred:
                continue
            total += 1
            g_best = max(gold, key=gold.get)
            p_best = max(pred, key=pred.get)
            correct += int(g_best == p_best)
        c["cats_correct"], c["cats_total"] = correct, total
    if "senter" in pipe_names:
        tp = fp = fn = 0
        for eg in examples:
            gs, ps = eg.reference.sent_starts, eg.predicted.sent_starts
            if gs is None or ps is None:
                continue
            for i in range(1, len(eg.reference)):  # position 0 is trivial
                g = int(gs[i]) > 0
                p = int(ps[i]) > 0
                tp += int(g and p)
                fp += int(p and not g)
                fn += int(g and not p)
        c["sent_tp"], c["sent_fp"], c["sent_fn"] = tp, fp, fn
    if "ner" in pipe_names or "entity_ruler" in pipe_names:
        tp = fp = fn = 0
        for eg in examples:
            gold = _ents_to_spans(eg.reference.ents or [])
            pred = _ents_to_spans(eg.predicted.ents or [])
            tp += len(gold & pred)
            fp += len(pred - gold)
            fn += len(gold - pred)
            # per-type counts (spaCy's ents_per_type), additive/mergeable
            for (_s, _e, lab) in gold & pred:
                c[f"ner_tp::{lab}"] = c.get(f"ner_tp::{lab}", 0) + 1
            for (_s, _e, lab) in pred - gold:
                c[f"ner_fp::{lab}"] = c.get(f"ner_fp::{lab}", 0) + 1
            for (_s, _e, lab) in gold - pred:
                c[f"ner_fn::{lab}"] = c.get(f"ner_fn::{lab}", 0) + 1
        c["ner_tp"], c["ner_fp"], c["ner_fn"] = tp, fp, fn
    return c


def merge_counts(parts: Sequence[Dict[str, int]]) -> Dict[str, int]:
    out: Dict[str, int] = {}
    for part in parts:
        for k, v in (part or {}).items():
            out[k] = out.get(k, 0) + v
    return out


def counts_to_scores(c: Dict[str, int]) -> Dict[str, float]:
    scores: Dict[str, float] = {}
    if "tag_total" in c:
        scores["tag_acc"] = c["tag_correct"] / c["tag_total"] if c["tag_total"] else 0.0
    if "dep_total" in c:
        t = c["dep_total"]
        scores["dep_uas"] = c["dep_uas_c"] / t if t else 0.0
        scores["dep_las"] = c["dep_las_c"] / t if t else 0.0
    if "lemma_total" in c:
        scores["lemma_acc"] = (c["lemma_correct"] / c["lemma_total"]
                               if c["lemma_total"] else 0.0)
    if "spans_tp" in c:
        tp, fp, fn = c["spans_tp"], c["spans_fp"], c["spans_fn"]
        p = tp / (tp + fp) if tp + fp else 0.0
        r = tp / (tp + fn) if tp + fn else 0.0
        scores["spans_sc_p"], scores["spans_sc_r"] = p, r
        scores["spans_sc_f"] = 2 * p * r / (p + r) if p + r else 0.0
    if "morph_total" in c:
        scores["morph_acc"] = (c["morph_correct"] / c["morph_total"]
                               if c["morph_total"] else 0.0)
    if "cats_total" in c:
        scores["cats_macro_acc"] = (c["cats_correct"] / c["cats_total"]
                                    if c["cats_total"] else 0.0)
    if "sent_tp" in c:
        tp, fp, fn = c["sent_tp"], c["sent_fp"], c["sent_fn"]
        p = tp / (tp + fp) if tp + fp else 0.0
        r = tp / (tp + fn) if tp + fn else 0.0
        scores["sents_p"], scores["sents_r"] = p, r
        scores["sents_f"] = 2 * p * r / (p + r) if p + r else 0.0
    if "ner_tp" in c:
        tp, fp, fn = c["ner_tp"], c["ner_fp"], c["ner_fn"]
        p = tp / (tp + fp) if tp + fp else 0.0
        r = tp / (tp + fn) if tp + fn else 0.0
        scores["ents_p"] = p
        scores["ents_r"] = r
        scores["ents_f"] = 2 * p * r / (p + r) if p + r else 0.0
        labels = {k.split("::", 1)[1] for k in c
                  if "::" in k and k.startswith("ner_")}
        if labels:
            per_type = {}
            for lab in sorted(labels):
                ltp = c.get(f"ner_tp::{lab}", 0)
                lfp = c.get(f"ner_fp::{lab}", 0)
                lfn = c.get(f"ner_fn::{lab}", 0)
                lp = ltp / (ltp + lfp) if ltp + lfp else 0.0
                lr = ltp / (ltp + lfn) if ltp + lfn else 0.0
                per_type[lab] = {"p": lp, "r": lr,
                                 "f": 2 * lp * lr / (lp + lr) if lp + lr else 0.0}
            scores["ents_per_type"] = per_type
    return scores


def score_examples(examples: Sequence[Example], pipe_names: Sequence[str]) -> Dict[str, float]:
    return counts_to_scores(score_counts(examples, pipe_names))


def weighted_score(scores: Dict[str, float], weights: Dict[str, float]) -> float:
    total = 0.0
    for key, w in (weights or {}).items():
        if w and key in scores and scores[key] is not None:
            total += w * scores[key]
    return total
