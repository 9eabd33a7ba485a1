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
