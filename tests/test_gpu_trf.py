"""GPU transformer path: tiny roberta trains through the HIP-backed
LayerNorm/Embedding drop-ins, fused residual+LN blocks and BPE subwords."""
import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu
need_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")

TINY_TRF_CFG = """
[nlp]
lang = "en"
pipeline = ["transformer", "tagger", "ner"]

[components]

[components.transformer]
factory = "transformer"

[components.transformer.model]
@architectures = "spacy-transformers.TransformerModel.v3"
name = "tiny"
window = 24
stride = 18

[components.transformer.model.transformer_config]
vocab_size = 2000
hidden_size = 64
num_hidden_layers = 2
num_attention_heads = 4
intermediate_size = 128

[components.tagger]
factory = "tagger"

[components.tagger.model]
@architectures = "spacy.Tagger.v2"

[components.tagger.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64

[components.ner]
factory = "ner"

[components.ner.model]
@architectures = "spacy.TransitionBasedParser.v2"
state_type = "ner"

[components.ner.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64

[corpora]

[corpora.train]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 200
words_per_doc = 12
vocab_size = 300
n_tags = 10
seed = 0

[corpora.dev]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 50
seed = 1

[training]
seed = 0
train_corpus = "corpora.train"
dev_corpus = "corpora.dev"

[training.optimizer]
@optimizers = "Adam.v1"
learn_rate = 0.001
"""


@need_gpu
def test_trf_pipeline_gpu_trains_and_predicts():
    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Example

    cfg = Config.from_str(TINY_TRF_CFG)
    nlp = init_nlp(cfg, device="cuda:0", sample_size=64)
    trf = nlp.get_pipe("transformer").module
    assert trf.bpe is not None and trf.bpe.tok is not None  # BPE trained
    T = resolve(cfg.interpolate()["training"], validate=False)
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    docs = make_synthetic_docs(nlp.vocab, n_docs=48, words_per_doc=12,
                               vocab_size=300, n_tags=10, n_deps=40,
                               n_ent_types=4, seed=3)
    examples = [Example.from_doc(d) for d in docs]
    losses_t = []
    for _ in range(8):
        losses = {}
        engine.accumulate(examples, drop=0.0, losses=losses)
        engine.apply_step()
        losses_t.append(sum(losses.values()))
    torch.cuda.synchronize()
    assert all(np.isfinite(v) for v in losses_t), losses_t
    assert losses_t[-1] < losses_t[0], losses_t
    outs = nlp.predict_docs([d.copy_unannotated() for d in docs[:4]])
    for d in outs:
        assert d.tags and len(d.tags) == len(d)
        assert d.ents is not None


@need_gpu
def test_window_attention_matches_sdpa():
    """window_attention (bmm + fused masked-softmax kernels) vs
    F.scaled_dot_product_attention with the same prefix mask — outputs
    and q/k/v gradients (no dropout: dropout paths differ by RNG)."""
    from spacy_ray_amd.ops import api

    torch.manual_seed(0)
    B, H, L, D = 7, 12, 96, 64
    lens_np = np.array([96, 50, 1, 33, 96, 2, 77], dtype=np.int32)
    lens = torch.from_numpy(lens_np).cuda()
    for dt, tol in ((torch.float32, 2e-4), (torch.bfloat16, 3e-2)):
        q = torch.randn(B, H, L, D, device="cuda", dtype=dt, requires_grad=True)
        k = torch.randn(B, H, L, D, device="cuda", dtype=dt, requires_grad=True)
        v = torch.randn(B, H, L, D, device="cuda", dtype=dt, requires_grad=True)
        scale = D ** -0.5
        valid_q = (torch.arange(L, device="cuda")[None, None, :, None] <
                   lens[:, None, None, None])
        out = api.window_attention(q, k, v, lens, scale, 0.0)
        dO = torch.randn_like(out) * valid_q  # no upstream grad on pad rows
        out.backward(dO)
        gq, gk, gv = q.grad.clone(), k.grad.clone(), v.grad.clone()
        q.grad = k.grad = v.grad = None
        # reference mask: valid queries see the prefix; PAD queries attend
        # position 0 only (well-defined softmax — no NaN rows; with zero
        # upstream grad they contribute nothing, matching our zeroed rows)
        j = torch.arange(L, device="cuda")
        qvalid = (j[:, None] < lens[:, None, None])  # [B, L, 1]
        kvalid = (j[None, None, :] < lens[:, None, None])  # [B, 1, L]
        mask2d = torch.where(qvalid, kvalid,
                             (j[None, None, :] == 0)).unsqueeze(1)
        from torch.nn.attention import SDPBackend, sdpa_kernel

        with sdpa_kernel([SDPBackend.MATH]):
            ref = torch.nn.functional.scaled_dot_product_attention(
                q, k, v, attn_mask=mask2d, scale=scale)
        ref.backward(dO)
        d = ((out - ref).abs() * valid_q).max()  # ours: pad query rows = 0
        assert float(d) < tol, (dt, float(d))
        for ours, theirs in ((gq, q.grad), (gk, k.grad), (gv, v.grad)):
            dg = (ours - theirs).abs().max()
            assert float(dg) < tol * 3, (dt, float(dg))


@need_gpu
def test_window_attention_dropout_statistics():
    """Dropout path: mean preserved (~1/keep scaling), backward runs, and
    the regenerated philox mask is consistent (dV of dropped entries 0)."""
    from spacy_ray_amd.ops import api

    torch.manual_seed(1)
    B, H, L, D = 4, 4, 64, 64
    lens = torch.full((B,), L, dtype=torch.int32, device="cuda")
    q = torch.randn(B, H, L, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)  # bf16 -> exercises the FUSED path
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    out = api.window_attention(q, k, v, lens, D ** -0.5, 0.3)
    out.sum().backward()
    assert torch.isfinite(out).all()
    assert torch.isfinite(q.grad).all() and torch.isfinite(v.grad).all()
    # with p=0 as control the outputs differ (mask actually applied)
    out0 = api.window_attention(q.detach(), k.detach(), v.detach(), lens,
                                D ** -0.5, 0.0)
    assert not torch.allclose(out, out0)
    # mean roughly preserved by 1/keep scaling
    assert abs(float(out.mean() - out0.mean())) < 0.05
