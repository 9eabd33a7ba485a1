"""Training loggers (@loggers registry).

Contract: ``logger(nlp) -> (print_row, finalize)`` — spaCy logger protocol
consumed at `/root/reference/spacy_ray/worker.py:190-193`.  Registered names:
  spacy.ConsoleLogger.v1      — standard table
  spacy-ray.ConsoleLogger.v1  — the reference's variant
    (`/root/reference/spacy_ray/loggers.py:8-66`): adds a wall-clock T column;
    the W column is the cluster-scaled word count.
  spacy-mi.JsonlLogger.v1     — metrics.jsonl sink (BASELINE.md reporting)
"""
from __future__ import annotations

import json
import time
from typing import Optional

from spacy_ray_amd.config.registry import registry


def _fmt(v, width=8):
    if v is None:
        return " " * width
    if isinstance(v, float):
        return f"{v:{width}.2f}"
    return f"{v:>{width}}"


def _make_console(nlp, with_time: bool):
    score_cols: list = []
    state = {"start": time.time(), "header": False}

    def print_row(info) -> None:
        loss_cols = sorted(info["losses"].keys())
        if not state["header"]:
            cols = (["T"] if with_time else []) + ["E", "#", "W"]
            cols += [f"Loss {c}" for c in loss_cols]
            sc = info.get("other_scores") or {}
            score_cols[:] = sorted(k for k, v in sc.items()
                                   if isinstance(v, (int, float)) and k != "speed")
            cols += score_cols + ["Score"]
            print("  ".join(f"{c:>10}" for c in cols))
            state["header"] = True
        row = []
        if with_time:
            row.append(_fmt(time.time() - state["start"], 10))
        row.append(_fmt(info["epoch"], 10))
        row.append(_fmt(info["step"], 10))
        row.append(_fmt(info.get("words_scaled", info["words"]), 10))
        for c in loss_cols:
            row.append(_fmt(info["losses"].get(c), 10))
        sc = info.get("other_scores") or {}
        for c in score_cols:
            v = sc.get(c)
            row.append(_fmt(100 * v if isinstance(v, float) else v, 10))
        s = info.get("score")
        row.append(_fmt(100 * s if isinstance(s, float) else s, 10))
        print("  ".join(row), flush=True)

    def finalize() -> None:
        pass

    return print_row, finalize


@registry.loggers("spacy.ConsoleLogger.v1")
def console_logger(progress_bar: bool = False):
    def setup(nlp):
        return _make_console(nlp, with_time=False)

    return setup


@registry.loggers("spacy-ray.ConsoleLogger.v1")
def ray_console_logger():
    def setup(nlp):
        return _make_console(nlp, with_time=True)

    return setup


@registry.loggers("spacy-mi.JsonlLogger.v1")
def jsonl_logger(path: Optional[str] = None, console: bool = True):
    def setup(nlp):
        print_console, finalize_console = _make_console(nlp, with_time=True)
        fh = open(path, "a") if path else None

        def print_row(info) -> None:
            if console:
                print_console(info)
            if fh is not None:
                rec = {
                    "step": info["step"],
                    "epoch": info["epoch"],
                    "losses": info["losses"],
                    "score": info.get("score"),
                    "other_scores": info.get("other_scores"),
                    "words": info["words"],
                    "words_seen": info.get("words_seen"),
                    "time": time.time(),
                }
                rec.update({k: info[k] for k in ("wps", "comm_ms", "compute_ms") if k in info})
                fh.write(json.dumps(rec) + "\n")
                fh.flush()

        def finalize() -> None:
            if fh is not None:
                fh.close()

        return print_row, finalize

    return setup
