"""GPU end-to-end: full pipeline trains on cuda:0 through the HIP kernels
(bf16 flat params) and the loss decreases; the HIP extension must actually
be the execution path (no silent torch fallback)."""
import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu
need_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")


@need_gpu
def test_hip_ext_required_on_gpu():
    """ops/api must route GPU tensors through _srx_hip (and would raise
    without it)."""
    assert os.environ.get("SRX_ALLOW_TORCH_FALLBACK") != "1"
    from spacy_ray_amd.ops import api

    assert api.hip_ext() is not None, "_srx_hip not importable on a GPU box"
    X = torch.randn(4, 3, 8, device="cuda")
    Y = api.maxout(X)
    assert Y.shape == (4, 8)


@need_gpu
def test_full_pipeline_gpu_train_loss_decreases():
    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Example

    cfg = Config.from_disk(os.path.join(os.path.dirname(__file__), "..",
                                        "examples", "configs", "en_core_cnn.cfg"))
    nlp = init_nlp(cfg, device="cuda:0", sample_size=32)
    T = resolve(cfg.interpolate()["training"], validate=False)
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    assert engine.flat_param.dtype == torch.bfloat16
    docs = make_synthetic_docs(nlp.vocab, n_docs=64, words_per_doc=15,
                               vocab_size=500, n_tags=50, n_deps=40,
                               n_ent_types=4, seed=11)
    examples = [Example.from_doc(d) for d in docs]
    losses_t = []
    for i in range(12):
        losses = {}
        engine.accumulate(examples, drop=0.0, losses=losses)
        engine.apply_step()
        losses_t.append(sum(losses.values()))
    torch.cuda.synchronize()
    assert all(np.isfinite(v) for v in losses_t), losses_t
    assert losses_t[-1] < losses_t[0], losses_t


@need_gpu
def test_gpu_predict_annotations():
    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.data.corpus import make_synthetic_docs

    cfg = Config.from_disk(os.path.join(os.path.dirname(__file__), "..",
                                        "examples", "configs", "en_core_cnn.cfg"))
    nlp = init_nlp(cfg, device="cuda:0", sample_size=16)
    docs = make_synthetic_docs(nlp.vocab, n_docs=4, words_per_doc=10,
                               vocab_size=200, n_tags=50, n_deps=40,
                               n_ent_types=4, seed=5)
    outs = nlp.predict_docs([d.copy_unannotated() for d in docs])
    for d in outs:
        assert d.tags and len(d.tags) == len(d)
        assert d.heads is not None and d.ents is not None


@need_gpu
def test_fused_step_parity(monkeypatch):
    """SRX_FUSED_STEP=1 (C++ autograd fused scorer+GEMM+select) must produce
    the same losses and gradients as the python per-step path (same kernels,
    same math; tolerance covers atomic-order + GEMM-solution jitter)."""
    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Example

    cfg = Config.from_disk(os.path.join(os.path.dirname(__file__), "..",
                                        "examples", "configs", "en_core_cnn.cfg"))

    def run(fused):
        monkeypatch.setenv("SRX_GPU_STATES", "0")  # exercise the step loop
        monkeypatch.setenv("SRX_FUSED_STEP", "1" if fused else "0")
        torch.manual_seed(0)
        np.random.seed(0)
        nlp = init_nlp(cfg, device="cuda:0", sample_size=32)
        T = resolve(cfg.interpolate()["training"], validate=False)
        engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
        docs = make_synthetic_docs(nlp.vocab, n_docs=48, words_per_doc=14,
                                   vocab_size=400, n_tags=50, n_deps=40,
                                   n_ent_types=4, seed=11)
        losses = {}
        engine.accumulate([Example.from_doc(d) for d in docs],
                          drop=0.0, losses=losses)
        torch.cuda.synchronize()
        return dict(losses), engine.grad_shard.float().clone()

    losses_ref, grad_ref = run(False)
    losses_fused, grad_fused = run(True)
    for k in losses_ref:
        assert abs(losses_fused[k] - losses_ref[k]) <= 1e-3 + 0.02 * abs(losses_ref[k]), (
            k, losses_ref[k], losses_fused[k])
    denom = grad_ref.abs().mean().clamp(min=1e-8)
    rel = (grad_fused - grad_ref).abs().mean() / denom
    assert float(rel) < 0.05, float(rel)


@need_gpu
def test_cpp_loop_parity(monkeypatch):
    """SRX_CPP_LOOP=1 (the C++-owned per-batch transition loop +
    batched CE/backward) must produce the same losses and gradients as the
    python round-robin path (same state machine, same scorer math;
    tolerance covers bf16 rounding + accumulation-order jitter)."""
    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Example

    cfg = Config.from_disk(os.path.join(os.path.dirname(__file__), "..",
                                        "examples", "configs", "en_core_cnn.cfg"))

    def run(cpp):
        monkeypatch.setenv("SRX_GPU_STATES", "0")  # exercise the host loops
        monkeypatch.setenv("SRX_CPP_LOOP", "1" if cpp else "0")
        torch.manual_seed(0)
        np.random.seed(0)
        nlp = init_nlp(cfg, device="cuda:0", sample_size=32)
        T = resolve(cfg.interpolate()["training"], validate=False)
        engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
        docs = make_synthetic_docs(nlp.vocab, n_docs=64, words_per_doc=14,
                                   vocab_size=400, n_tags=50, n_deps=40,
                                   n_ent_types=4, seed=13)
        losses = {}
        engine.accumulate([Example.from_doc(d) for d in docs],
                          drop=0.0, losses=losses)
        torch.cuda.synchronize()
        # decode parity on the same docs
        pred = [d.copy_unannotated() for d in docs]
        nlp.predict_docs(pred)
        heads = np.concatenate([d.heads for d in pred])
        tags = np.concatenate([np.asarray(d.ents, dtype=object) for d in pred])
        return dict(losses), engine.grad_shard.float().clone(), heads, tags

    losses_ref, grad_ref, heads_ref, ents_ref = run(False)
    losses_cpp, grad_cpp, heads_cpp, ents_cpp = run(True)
    for k in losses_ref:
        assert abs(losses_cpp[k] - losses_ref[k]) <= 1e-3 + 0.02 * abs(losses_ref[k]), (
            k, losses_ref[k], losses_cpp[k])
    denom = grad_ref.abs().mean().clamp(min=1e-8)
    rel = (grad_cpp - grad_ref).abs().mean() / denom
    assert float(rel) < 0.05, float(rel)
    # greedy decode: near-ties under bf16 rounding may flip a few actions
    assert (heads_cpp == heads_ref).mean() > 0.97
    assert (ents_cpp == ents_ref).mean() > 0.97


@need_gpu
def test_gpu_state_machine_parity(monkeypatch):
    """SRX_GPU_STATES=1 (fully GPU-resident transition loop: one wave per
    doc runs the whole greedy parse/NER in-kernel, srx_gpustate.hip) must
    match the C++ host loop: same losses, same gradients, same decode
    annotations (same state machine + oracle + scorer math re-derived
    in-kernel; tolerance covers bf16 accumulation-order jitter)."""
    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Example

    cfg = Config.from_disk(os.path.join(os.path.dirname(__file__), "..",
                                        "examples", "configs", "en_core_cnn.cfg"))

    def run(gpu_states):
        monkeypatch.setenv("SRX_GPU_STATES", "1" if gpu_states else "0")
        torch.manual_seed(0)
        np.random.seed(0)
        nlp = init_nlp(cfg, device="cuda:0", sample_size=32)
        T = resolve(cfg.interpolate()["training"], validate=False)
        engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
        docs = make_synthetic_docs(nlp.vocab, n_docs=64, words_per_doc=14,
                                   vocab_size=400, n_tags=50, n_deps=40,
                                   n_ent_types=4, seed=13)
        examples = [Example.from_doc(d) for d in docs]
        losses = {}
        engine.accumulate(examples, drop=0.0, losses=losses)
        torch.cuda.synchronize()
        pred = [d.copy_unannotated() for d in docs]
        nlp.predict_docs(pred)
        heads = np.concatenate([d.heads for d in pred])
        deps = np.concatenate([np.asarray(d.deps, dtype=object) for d in pred])
        ents = np.concatenate([np.asarray(d.ents, dtype=object) for d in pred])
        return dict(losses), engine.grad_shard.float().clone(), heads, deps, ents

    losses_ref, grad_ref, heads_ref, deps_ref, ents_ref = run(False)
    losses_gpu, grad_gpu, heads_gpu, deps_gpu, ents_gpu = run(True)
    for k in losses_ref:
        assert abs(losses_gpu[k] - losses_ref[k]) <= 1e-3 + 0.02 * abs(losses_ref[k]), (
            k, losses_ref[k], losses_gpu[k])
    denom = grad_ref.abs().mean().clamp(min=1e-8)
    rel = (grad_gpu - grad_ref).abs().mean() / denom
    assert float(rel) < 0.05, float(rel)
    # greedy decode: near-ties under bf16 rounding may flip a few actions
    assert (heads_gpu == heads_ref).mean() > 0.97
    assert (deps_gpu == deps_ref).mean() > 0.97
    assert (ents_gpu == ents_ref).mean() > 0.97


@need_gpu
def test_gpu_state_machine_long_doc_fallback(monkeypatch):
    """Docs above GPU_STATE_MAXLEN must fall back to the host loop (mixed
    batch still trains and decodes)."""
    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Example

    monkeypatch.setenv("SRX_GPU_STATES", "1")
    cfg = Config.from_disk(os.path.join(os.path.dirname(__file__), "..",
                                        "examples", "configs", "en_core_cnn.cfg"))
    nlp = init_nlp(cfg, device="cuda:0", sample_size=16)
    T = resolve(cfg.interpolate()["training"], validate=False)
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    docs = make_synthetic_docs(nlp.vocab, n_docs=8, words_per_doc=200,
                               vocab_size=300, n_tags=50, n_deps=40,
                               n_ent_types=4, seed=3)
    assert max(len(d) for d in docs) > 128  # forces the parser fallback
    losses = {}
    engine.accumulate([Example.from_doc(d) for d in docs], drop=0.0,
                      losses=losses)
    engine.apply_step()
    torch.cuda.synchronize()
    assert all(np.isfinite(v) for v in losses.values()), losses
    pred = [d.copy_unannotated() for d in docs]
    nlp.predict_docs(pred)
    for d in pred:
        assert d.heads is not None and len(d.heads) == len(d)
        assert d.ents is not None


@need_gpu
def test_transition_ce_kernel_matches_torch():
    """transition_ce (fused masked softmax CE + dScores) vs the torch
    composition used by the python path."""
    from spacy_ray_amd.ops.api import hip_ext

    hip = hip_ext()
    torch.manual_seed(3)
    SS, A = 512, 83
    scores = torch.randn(SS, A, device="cuda", dtype=torch.float32)
    valid = (torch.rand(SS, A, device="cuda") < 0.6).to(torch.uint8)
    valid[:, 0] = 1  # every row has a valid action
    gold = ((torch.rand(SS, A, device="cuda") < 0.2).to(torch.uint8) & valid)
    gold[::7] = 0  # some unsupervised rows
    loss_count, dScores, colsum = hip.transition_ce(scores, gold, valid)
    # torch reference
    NEG_INF = -1e30
    g = gold > 0
    v = valid > 0
    counts = g.sum(-1)
    ok = counts > 0
    logp = torch.log_softmax(scores.masked_fill(~v, NEG_INF), dim=-1)
    target = g.float() / counts.clamp(min=1).unsqueeze(-1)
    row_loss = -(target * logp).sum(-1)
    ref_loss = row_loss.masked_fill(~ok, 0).sum()
    ref_d = (logp.exp() * v.float() - target) * ok.unsqueeze(-1).float()
    ref_d = ref_d * v.float()
    assert torch.allclose(loss_count[0], ref_loss, rtol=1e-3, atol=1e-3)
    assert float(loss_count[1]) == float(ok.sum())
    assert torch.allclose(dScores, ref_d, rtol=1e-3, atol=1e-4)
    assert torch.allclose(colsum, ref_d.sum(0), rtol=1e-3, atol=1e-3)


@need_gpu
def test_dpre_scatter_matches_index_add():
    from spacy_ray_amd.ops.api import hip_ext

    hip = hip_ext()
    torch.manual_seed(4)
    SS, nF, HP, T1 = 700, 13, 128, 300
    pad = T1 - 1
    dSummed = torch.randn(SS, HP, device="cuda", dtype=torch.float32)
    feats = torch.randint(0, T1, (SS, nF), device="cuda", dtype=torch.int64)
    out = torch.zeros(T1, nF, HP, device="cuda", dtype=torch.float32)
    dBias, dPad = hip.dpre_scatter(dSummed, feats, out, pad)
    # reference: index_add over non-pad entries; pad sums come back in dPad
    ref = torch.zeros(T1 * nF, HP, device="cuda", dtype=torch.float32)
    slot = torch.arange(nF, device="cuda")
    dest = (feats * nF + slot).reshape(-1)
    src = dSummed.repeat_interleave(nF, dim=0)
    mask = (feats != pad).reshape(-1)
    ref.index_add_(0, dest[mask], src[mask])
    assert torch.allclose(out.view(T1 * nF, HP), ref, rtol=1e-4, atol=1e-3)
    assert torch.allclose(dBias, dSummed.sum(0), rtol=1e-4, atol=1e-3)
    ref_pad = torch.zeros(nF, HP, device="cuda")
    for f in range(nF):
        ref_pad[f] = dSummed[feats[:, f] == pad].sum(0)
    assert torch.allclose(dPad, ref_pad, rtol=1e-4, atol=1e-3)
    # bf16 output path (packed atomics): looser tolerance, same structure
    dS16 = dSummed.to(torch.bfloat16)
    out16 = torch.zeros(T1, nF, HP, device="cuda", dtype=torch.bfloat16)
    dBias16, dPad16 = hip.dpre_scatter(dS16, feats, out16, pad)
    ref16 = torch.zeros(T1 * nF, HP, device="cuda", dtype=torch.float32)
    ref16.index_add_(0, dest[mask], dS16.float().repeat_interleave(nF, dim=0)[mask])
    diff = (out16.float().view(T1 * nF, HP) - ref16).abs()
    assert float(diff.mean()) < 0.05, float(diff.mean())
    assert torch.allclose(dBias16, dS16.float().sum(0), rtol=1e-2, atol=0.5)


@need_gpu
def test_nccl_world1_fused_collectives(monkeypatch):
    """The REAL RCCL path (`reduce_scatter_tensor`/`all_gather_into_tensor`
    on backend nccl, DistComm._fused branch) executes for a full training
    step — world=1 because RCCL refuses two ranks on one device ('Duplicate
    GPU detected', probed on MI355X); the driver's multi-GPU SCALE run is
    the world>1 exercise.  SRX_COLLECTIVE_CHECK asserts the op sequence."""
    import torch.distributed as dist

    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import DistComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Example

    monkeypatch.setenv("MASTER_ADDR", "127.0.0.1")
    monkeypatch.setenv("MASTER_PORT", "29617")
    monkeypatch.setenv("RANK", "0")
    monkeypatch.setenv("WORLD_SIZE", "1")
    monkeypatch.setenv("SRX_COLLECTIVE_CHECK", "1")
    assert not dist.is_initialized()
    try:
        comm = DistComm(backend="nccl", device=torch.device("cuda:0"))
        assert comm._fused
        cfg = Config.from_disk(os.path.join(os.path.dirname(__file__), "..",
                                            "examples", "configs", "en_core_cnn.cfg"))
        nlp = init_nlp(cfg, device="cuda:0", sample_size=16)
        T = resolve(cfg.interpolate()["training"], validate=False)
        engine = ZeRO1Engine(nlp, T["optimizer"], comm)
        docs = make_synthetic_docs(nlp.vocab, n_docs=16, words_per_doc=12,
                                   vocab_size=300, n_tags=50, n_deps=40,
                                   n_ent_types=4, seed=5)
        losses = {}
        for _ in range(2):
            engine.accumulate([Example.from_doc(d) for d in docs],
                              drop=0.0, losses=losses)
            engine.apply_step()
        comm.barrier()  # runs the collective-sequence check
        torch.cuda.synchronize()
        assert comm._oplog == []  # consumed by the check
        assert all(v >= 0 for v in losses.values())
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


@need_gpu
def test_deterministic_mode_bit_identical_grads(monkeypatch):
    """SRX_DETERMINISTIC=1: two identical runs produce BIT-IDENTICAL flat
    gradients (fixed-point int64 atomics + stable sorts; SURVEY §5.2,
    VERDICT r1 item 7)."""
    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Example

    monkeypatch.setenv("SRX_DETERMINISTIC", "1")
    cfg = Config.from_disk(os.path.join(os.path.dirname(__file__), "..",
                                        "examples", "configs", "en_core_cnn.cfg"))

    def run():
        torch.manual_seed(0)
        np.random.seed(0)
        nlp = init_nlp(cfg, device="cuda:0", sample_size=32)
        T = resolve(cfg.interpolate()["training"], validate=False)
        engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
        docs = make_synthetic_docs(nlp.vocab, n_docs=96, words_per_doc=16,
                                   vocab_size=600, n_tags=50, n_deps=40,
                                   n_ent_types=4, seed=17)
        engine.accumulate([Example.from_doc(d) for d in docs], drop=0.0)
        torch.cuda.synchronize()
        return engine.grad_shard.clone()

    g1 = run()
    g2 = run()
    assert torch.equal(g1, g2), float((g1 - g2).abs().max())
    # sanity: WITHOUT the flag the grads are still statistically equal but
    # this assertion documents that det mode is what guarantees bits
    monkeypatch.delenv("SRX_DETERMINISTIC")
    g3 = run()
    rel = (g3.float() - g1.float()).abs().mean() / g1.float().abs().mean().clamp(min=1e-8)
    assert float(rel) < 0.02


@need_gpu
def test_textcat_senter_gpu_train():
    """textcat + senter pipes on GPU: kernels behind linear_cdw /
    reduce_mean_ragged carry doc classification and sentence recognition;
    losses fall."""
    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Example

    cfg = Config.from_disk(os.path.join(os.path.dirname(__file__), "..",
                                        "examples", "configs",
                                        "en_textcat.cfg"))
    nlp = init_nlp(cfg, device="cuda:0", sample_size=64)
    T = resolve(cfg.interpolate()["training"], validate=False)
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    docs = make_synthetic_docs(nlp.vocab, n_docs=96, words_per_doc=16,
                               vocab_size=400, n_tags=50, n_deps=40,
                               n_ent_types=4, seed=9)
    examples = [Example.from_doc(d) for d in docs]
    first = last = None
    for i in range(12):
        losses = {}
        engine.accumulate(examples, drop=0.0, losses=losses)
        engine.apply_step()
        if i == 0:
            first = dict(losses)
        last = dict(losses)
    torch.cuda.synchronize()
    assert last["textcat"] < first["textcat"], (first, last)
    assert last["senter"] < first["senter"], (first, last)
    pred = [d.copy_unannotated() for d in docs[:8]]
    nlp.predict_docs(pred)
    for d in pred:
        assert d.cats and d.sent_starts is not None


@need_gpu
def test_gpu_state_machine_exact_maxlen_boundary(monkeypatch):
    """Docs of exactly GPU_STATE_MAXLEN (128) tokens run on the GPU state
    machine (the LDS arrays/bitmaps are sized for exactly this bound) and
    match the host loop; 129-token docs fall back."""
    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Doc, Example

    cfg = Config.from_disk(os.path.join(os.path.dirname(__file__), "..",
                                        "examples", "configs", "en_core_cnn.cfg"))

    def make_docs(nlp):
        base = make_synthetic_docs(nlp.vocab, n_docs=8, words_per_doc=40,
                                   vocab_size=300, n_tags=50, n_deps=40,
                                   n_ent_types=4, seed=21)
        docs = []
        for want in (127, 128, 128, 1, 2):
            d0 = base[len(docs) % len(base)]
            reps = (want + len(d0) - 1) // len(d0)
            words = (d0.words * reps)[:want]
            tags = (d0.tags * reps)[:want]
            heads = [max(0, i - 1) for i in range(want)]
            heads[0] = -1
            deps = ["dep0"] * want
            deps[0] = "ROOT"
            ents = (d0.ents * reps)[:want]
            # re-open truncated spans safely: strip a leading I/L
            for i in (0,):
                if ents[i][0] in "IL":
                    ents[i] = "O"
            docs.append(Doc(nlp.vocab, words, tags=tags, heads=heads,
                            deps=deps, ents=["O"] * want))
        return docs

    def run(gpu_states):
        monkeypatch.setenv("SRX_GPU_STATES", "1" if gpu_states else "0")
        torch.manual_seed(0)
        np.random.seed(0)
        nlp = init_nlp(cfg, device="cuda:0", sample_size=16)
        T = resolve(cfg.interpolate()["training"], validate=False)
        engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
        docs = make_docs(nlp)
        examples = [Example.from_doc(d) for d in docs]
        losses = {}
        engine.accumulate(examples, drop=0.0, losses=losses)
        torch.cuda.synchronize()
        pred = [d.copy_unannotated() for d in docs]
        nlp.predict_docs(pred)
        heads = np.concatenate([d.heads for d in pred])
        return dict(losses), heads

    losses_ref, heads_ref = run(False)
    losses_gpu, heads_gpu = run(True)
    for k in losses_ref:
        assert abs(losses_gpu[k] - losses_ref[k]) <= 1e-3 + 0.03 * abs(losses_ref[k]), (
            k, losses_ref[k], losses_gpu[k])
    assert (heads_gpu == heads_ref).mean() > 0.97
