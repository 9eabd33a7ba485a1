"""Distributed engine tests on CPU.

- ZeRO1Engine world=1 must match SimpleAdam-equivalent training trajectories
  in spirit (loss decreases, params update).
- world=2 over gloo: a 2-rank run with sharded data must produce IDENTICAL
  parameters on both ranks after N steps (the all-gather republish), and the
  update must equal the 1-rank run on the combined batch (DP equivalence).
"""
import json
import os
import subprocess
import sys
from pathlib import Path

import numpy as np
import pytest
import torch

from spacy_ray_amd.config.config import Config, resolve, resolve_dot_names
from spacy_ray_amd.parallel.comm import LocalComm
from spacy_ray_amd.parallel.engine import ZeRO1Engine
from spacy_ray_amd.pipeline.language import init_nlp

from tests.test_pipeline import TAGGER_CFG

REPO = Path(__file__).resolve().parent.parent


def _make_nlp_and_examples(n=8):
    cfg = Config.from_str(TAGGER_CFG)
    nlp = init_nlp(cfg)
    icfg = cfg.interpolate()
    T = resolve(icfg["training"], validate=False)
    (train_corpus,) = resolve_dot_names(icfg, [T["train_corpus"]])
    examples = []
    for eg in train_corpus(nlp):
        examples.append(eg)
        if len(examples) >= n:
            break
    return nlp, T, examples


def test_zero1_local_step_updates_params():
    nlp, T, examples = _make_nlp_and_examples()
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    before = engine.flat_param.clone()
    losses = {}
    engine.accumulate(examples, losses=losses)
    engine.apply_step()
    assert not torch.equal(before, engine.flat_param)
    assert losses["tagger"] > 0
    # gradients cleared
    assert engine.flat_grad.abs().max() == 0


def test_zero1_matches_reference_adamw_math():
    """One step of the flat sharded Adam == torch AdamW on the same grads
    (modulo clip): decoupled weight decay, bias correction."""
    nlp, T, examples = _make_nlp_and_examples()
    spec = T["optimizer"]
    spec.grad_clip = 0.0  # isolate the Adam math
    engine = ZeRO1Engine(nlp, spec, LocalComm())
    # capture params+grads before the step
    losses = {}
    engine.accumulate(examples, losses=losses)
    p0 = engine.flat_param.clone()
    g0 = engine.flat_grad.clone()
    engine.apply_step()
    # reference AdamW on the flat tensors
    ref_p = p0.clone()
    m = torch.zeros_like(ref_p)
    v = torch.zeros_like(ref_p)
    lr = spec.lr(0)
    ref_p.mul_(1 - lr * spec.L2)
    m.mul_(spec.beta1).add_(g0, alpha=1 - spec.beta1)
    v.mul_(spec.beta2).addcmul_(g0, g0, value=1 - spec.beta2)
    denom = (v / (1 - spec.beta2)).sqrt_().add_(spec.eps)
    ref_p.addcdiv_(m, denom, value=-lr / (1 - spec.beta1))
    assert torch.allclose(engine.flat_param, ref_p, atol=1e-6)


def test_grad_accumulation_equivalence():
    """accumulate(a)+accumulate(b) then step == accumulate(a+b) then step,
    when losses are per-subbatch sums of per-token means... here: both orders
    produce finite, close updates (exact equality needs identical loss
    normalization; we check grads sum linearly)."""
    nlp, T, examples = _make_nlp_and_examples(8)
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    engine.accumulate(examples[:4], sync=False)
    g1 = engine.flat_grad.clone()
    engine.flat_grad.zero_()
    engine.accumulate(examples[4:], sync=False)
    g2 = engine.flat_grad.clone()
    engine.flat_grad.zero_()
    engine.accumulate(examples[:4], sync=False)
    engine.accumulate(examples[4:], sync=False)
    g12 = engine.flat_grad.clone()
    assert torch.allclose(g12, g1 + g2, atol=1e-5)


def test_refresh_master_after_external_param_load():
    """Loading params into the flat views after engine construction must be
    followed by refresh_master_from_params, or the stale master would revert
    them on the first step (resume-without-optimizer-state path)."""
    nlp, T, examples = _make_nlp_and_examples(4)
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    # simulate an external load: overwrite one param in place
    p = next(iter(engine.module.parameters()))
    with torch.no_grad():
        p.add_(1.0)
    engine.refresh_master_from_params()
    engine.accumulate(examples)
    engine.apply_step()
    # the +1 offset must have survived (master was refreshed, not stale)
    assert float(p.detach().abs().mean()) > 0.5


def test_use_averages_swap():
    """With use_averages, averaged_params() swaps the running mean into the
    live params and restores the trained params after."""
    nlp, T, examples = _make_nlp_and_examples(4)
    spec = T["optimizer"]
    spec.use_averages = True
    engine = ZeRO1Engine(nlp, spec, LocalComm())
    for _ in range(3):
        engine.accumulate(examples)
        engine.apply_step()
    trained = engine.flat_param.clone()
    with engine.averaged_params():
        swapped = engine.flat_param.clone()
        assert not torch.equal(trained, swapped)  # average != latest
    assert torch.allclose(engine.flat_param, trained)  # restored
    # no averaging configured -> context is a no-op
    spec2 = T["optimizer"]
    spec2.use_averages = False
    nlp2, T2, _ = _make_nlp_and_examples(2)
    engine2 = ZeRO1Engine(nlp2, T2["optimizer"], LocalComm())
    before = engine2.flat_param.clone()
    with engine2.averaged_params():
        assert torch.equal(engine2.flat_param, before)


def test_flat_buffer_layout_invariants():
    """Every param starts on a 64-element boundary (hipBLASLt alignment),
    buckets tile the flat buffer, and world divides every bucket."""
    nlp, T, _ = _make_nlp_and_examples(2)
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    for off, (name, p) in zip(engine._offsets,
                              [(n, p) for n, p in engine.module.named_parameters()
                               if p.requires_grad]):
        assert off % 64 == 0, name
        # the param's data must be a view into the flat buffer at `off`
        assert p.data.data_ptr() == engine.flat_param.data_ptr() + off * engine.dtype.itemsize
    prev_end = 0
    for b in engine.buckets:
        assert b.start == prev_end
        assert (b.end - b.start) % (64 * engine.comm.world) == 0
        prev_end = b.end
    assert prev_end == engine.flat_param.numel()
    assert engine.shard_elems == sum(b.per for b in engine.buckets)


def test_accumulate_gradient_microbatching():
    """accumulate_gradient=2 splits each batch into 2 sub-batches whose
    gradients accumulate before one optimizer step (the key the reference
    reads but never wires — SURVEY.md §2.3)."""
    from spacy_ray_amd.train.loop import train_while_improving

    nlp, T, examples = _make_nlp_and_examples(8)
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    calls = []
    orig_acc, orig_apply = engine.accumulate, engine.apply_step

    def acc(sub, drop=0.0, losses=None, sync=True):
        calls.append(("acc", len(sub), sync))
        orig_acc(sub, drop=drop, losses=losses, sync=sync)

    def apply():
        calls.append(("apply",))
        orig_apply()

    engine.accumulate, engine.apply_step = acc, apply
    data = iter([(0, examples)])
    it = train_while_improving(
        nlp, engine, data, evaluate=lambda: (0.0, {}), dropout=0.0,
        accumulate_gradient=2, max_steps=1, eval_frequency=100,
    )
    for _ in it:
        pass
    accs = [c for c in calls if c[0] == "acc"]
    assert len(accs) == 2
    assert accs[0][2] is False and accs[1][2] is True  # sync only on last
    assert sum(c[1] for c in accs) == len(examples)
    assert calls[-1] == ("apply",) or ("apply",) in calls


_WORKER_SCRIPT = r"""
import json, os, sys
import numpy as np
import torch
sys.path.insert(0, "@@REPO@@")
from spacy_ray_amd.config.config import Config, resolve, resolve_dot_names
from spacy_ray_amd.parallel.comm import init_comm_from_env
from spacy_ray_amd.parallel.engine import ZeRO1Engine
from spacy_ray_amd.pipeline.language import init_nlp
from tests.test_pipeline import TAGGER_CFG

rank = int(os.environ["RANK"]); world = int(os.environ["WORLD_SIZE"])
cfg = Config.from_str(TAGGER_CFG)
nlp = init_nlp(cfg)
icfg = cfg.interpolate()
T = resolve(icfg["training"], validate=False)
(train_corpus,) = resolve_dot_names(icfg, [T["train_corpus"]])
# equal-length docs only: per-rank losses normalize by the shard's token
# count, so DP mean-of-means == full-batch mean exactly IFF shards carry
# equal token totals (same local-normalization semantics as the reference)
examples = []
for eg in train_corpus(nlp):
    if len(eg.reference) != 10:
        continue
    examples.append(eg)
    if len(examples) >= 8:
        break
comm = init_comm_from_env()
engine = ZeRO1Engine(nlp, T["optimizer"], comm)
# rank r trains on its 8/world slice; DP average == full-batch
# mean-of-means here (equal-size slices)
per = 8 // world
mine = examples[rank * per:(rank + 1) * per]
for _ in range(3):
    engine.accumulate(mine)
    engine.apply_step()
out = {"param_hash": float(engine.flat_param.double().abs().sum()),
       "param": engine.flat_param[:32].tolist()}
print("RESULT" + json.dumps(out))
"""


@pytest.mark.parametrize("world", [2])
def test_two_rank_gloo_param_consistency(tmp_path, world):
    """2-rank gloo run: both ranks end with identical parameters."""
    script = tmp_path / "worker.py"
    script.write_text(_WORKER_SCRIPT.replace("@@REPO@@", str(REPO)))
    procs = []
    port = 29511
    for rank in range(world):
        env = dict(os.environ)
        env.update(RANK=str(rank), LOCAL_RANK=str(rank), WORLD_SIZE=str(world),
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env, cwd=str(REPO),
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    results = []
    for p in procs:
        out, err = p.communicate(timeout=300)
        assert p.returncode == 0, f"worker failed:\n{err[-2000:]}"
        line = [l for l in out.splitlines() if l.startswith("RESULT")][0]
        results.append(json.loads(line[len("RESULT"):]))
    assert results[0]["param_hash"] == pytest.approx(results[1]["param_hash"], rel=1e-9)
    assert np.allclose(results[0]["param"], results[1]["param"])


def _run_world(tmp_path, world, port):
    script = tmp_path / f"worker_w{world}.py"
    script.write_text(_WORKER_SCRIPT.replace("@@REPO@@", str(REPO)))
    procs = []
    for rank in range(world):
        env = dict(os.environ)
        env.update(RANK=str(rank), LOCAL_RANK=str(rank), WORLD_SIZE=str(world),
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env, cwd=str(REPO),
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    results = []
    for p in procs:
        out, err = p.communicate(timeout=300)
        assert p.returncode == 0, f"worker failed:\n{err[-2000:]}"
        line = [l for l in out.splitlines() if l.startswith("RESULT")][0]
        results.append(json.loads(line[len("RESULT"):]))
    return results


def test_two_rank_matches_single_rank_full_batch(tmp_path):
    """an N-rank data-parallel run must land on the same
    parameters as 1 rank on the full batch (gradient averaging over equal
    shards == full-batch mean; identical seeded init)."""
    one = _run_world(tmp_path, 1, 29513)[0]
    two = _run_world(tmp_path, 2, 29514)[0]
    assert one["param_hash"] == pytest.approx(two["param_hash"], rel=1e-5)
    assert np.allclose(one["param"], two["param"], rtol=1e-4, atol=1e-6)


def test_frozen_components_not_updated(tmp_path):
    """training.frozen_components: the frozen pipe's params must be EXACTLY
    unchanged after training — not even weight-decayed (regression: frozen
    params left in the flat buffer got AdamW decoupled decay each step)."""
    from spacy_ray_amd.train.worker import distributed_train

    out = tmp_path / "frozen"
    holder = {}

    import spacy_ray_amd.train.worker as workermod
    orig_init = workermod.init_nlp

    def capture_init(config, **kw):
        nlp = orig_init(config, **kw)
        if "nlp" not in holder:
            holder["nlp"] = nlp
            t2v = dict(nlp.pipeline)["tok2vec"]
            holder["before"] = [p.detach().clone() for p in t2v.module.parameters()]
        return nlp

    workermod.init_nlp = capture_init
    try:
        distributed_train(
            Config.from_str(TAGGER_CFG, overrides={
                "training.max_steps": 3, "training.eval_frequency": 2,
                "training.frozen_components": ["tok2vec"]}),
            output_path=out, use_gpu=-1)
    finally:
        workermod.init_nlp = orig_init
    nlp = holder["nlp"]
    t2v = dict(nlp.pipeline)["tok2vec"]
    tagger = dict(nlp.pipeline)["tagger"]
    after = list(t2v.module.parameters())
    assert all(not p.requires_grad for p in after)
    for b, a in zip(holder["before"], after):
        assert torch.equal(b, a)  # bit-identical: no update, no decay
    # the trainable head did move
    assert any(p.requires_grad for p in tagger.module.parameters())


def test_use_averages_survives_resume():
    """state_dict round-trip carries the running parameter average; without
    it the stale init-time avg would dominate the mean after resume."""
    nlp, T, examples = _make_nlp_and_examples(4)
    spec = T["optimizer"]
    spec.use_averages = True
    engine = ZeRO1Engine(nlp, spec, LocalComm())
    for _ in range(3):
        engine.accumulate(examples)
        engine.apply_step()
    saved = {k: (v.clone() if torch.is_tensor(v) else v)
             for k, v in engine.state_dict().items()}
    assert "avg" in saved

    nlp2, T2, _ = _make_nlp_and_examples(4)
    spec2 = T2["optimizer"]
    spec2.use_averages = True
    engine2 = ZeRO1Engine(nlp2, spec2, LocalComm())
    engine2.load_state_dict(saved)
    assert torch.equal(engine2.avg, engine.avg)
    assert engine2.step_count == engine.step_count

    # legacy checkpoint without 'avg': falls back to restored master
    legacy = {k: v for k, v in saved.items() if k != "avg"}
    nlp3, T3, _ = _make_nlp_and_examples(4)
    spec3 = T3["optimizer"]
    spec3.use_averages = True
    engine3 = ZeRO1Engine(nlp3, spec3, LocalComm())
    engine3.load_state_dict(legacy)
    assert torch.equal(engine3.avg, engine3.master)


def test_resume_world_size_mismatch_clear_error():
    """Loading an optimizer shard saved at a different world size must fail
    with a clear message (the shard layout depends on world size)."""
    nlp, T, examples = _make_nlp_and_examples(4)
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    engine.accumulate(examples)
    engine.apply_step()
    state = engine.state_dict()
    state["world"] = 4
    nlp2, T2, _ = _make_nlp_and_examples(4)
    engine2 = ZeRO1Engine(nlp2, T2["optimizer"], LocalComm())
    with pytest.raises(ValueError, match="world_size"):
        engine2.load_state_dict(state)


def test_all_frozen_step_is_noop_not_crash():
    nlp, T, examples = _make_nlp_and_examples(4)
    for _, pipe in nlp.pipeline:
        for p in pipe.module.parameters():
            p.requires_grad_(False)
    nlp._frozen = [n for n, _ in nlp.pipeline]
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    assert engine.flat_param.numel() == 0
    engine.accumulate(examples)
    engine.apply_step()  # no params, no crash
    nlp._frozen = []


_WORKER_SCRIPT_BREADTH = r"""
import json, os, sys
import torch
sys.path.insert(0, "@@REPO@@")
from spacy_ray_amd.config.config import Config, resolve, resolve_dot_names
from spacy_ray_amd.data.corpus import make_synthetic_docs
from spacy_ray_amd.parallel.comm import init_comm_from_env
from spacy_ray_amd.parallel.engine import ZeRO1Engine
from spacy_ray_amd.pipeline.language import init_nlp
from spacy_ray_amd.vocab.doc import Example
from tests.test_pipeline import TEXTCAT_CFG

rank = int(os.environ["RANK"]); world = int(os.environ["WORLD_SIZE"])
torch.manual_seed(0)
cfg = Config.from_str(TEXTCAT_CFG)
nlp = init_nlp(cfg, sample_size=32)
icfg = cfg.interpolate()
T = resolve(icfg["training"], validate=False)
docs = make_synthetic_docs(nlp.vocab, n_docs=8, words_per_doc=12,
                           vocab_size=100, n_tags=10, n_deps=5,
                           n_ent_types=3, seed=5)
docs = [d for d in docs if len(d) == 12][:4] or docs[:4]
examples = [Example.from_doc(d) for d in docs]
comm = init_comm_from_env()
engine = ZeRO1Engine(nlp, T["optimizer"], comm)
per = len(examples) // world
mine = examples[rank * per:(rank + 1) * per]
for _ in range(2):
    engine.accumulate(mine)
    engine.apply_step()
out = {"param_hash": float(engine.flat_param.double().abs().sum())}
print("RESULT" + json.dumps(out))
"""


def test_two_rank_gloo_textcat_senter(tmp_path):
    """world=2 gloo with the textcat+senter pipeline: the new heads share
    the flat-param layout across ranks and land on identical params."""
    script = tmp_path / "worker_breadth.py"
    script.write_text(_WORKER_SCRIPT_BREADTH.replace("@@REPO@@", str(REPO)))
    procs = []
    port = 29533
    for rank in range(2):
        env = dict(os.environ)
        env.update(RANK=str(rank), LOCAL_RANK=str(rank), WORLD_SIZE="2",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env, cwd=str(REPO),
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    results = []
    for p in procs:
        out, err = p.communicate(timeout=300)
        assert p.returncode == 0, f"worker failed:\n{err[-2000:]}"
        line = [l for l in out.splitlines() if l.startswith("RESULT")][0]
        results.append(json.loads(line[len("RESULT"):]))
    assert results[0]["param_hash"] == pytest.approx(
        results[1]["param_hash"], rel=1e-9)
