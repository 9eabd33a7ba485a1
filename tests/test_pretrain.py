"""Tok2vec pretraining (spaCy `pretrain` role): masked-token objective,
weight save/load through `training.init_tok2vec`."""
import numpy as np
import torch

from spacy_ray_amd.config.config import Config
from spacy_ray_amd.pipeline.language import init_nlp
from spacy_ray_amd.train.pretrain import (load_init_tok2vec, pretrain_tok2vec,
                                          save_tok2vec)

CFG = "examples/configs/en_tagger_cpu.cfg"


def _corpus(nlp):
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Example

    docs = make_synthetic_docs(nlp.vocab, n_docs=96, words_per_doc=12,
                               vocab_size=120, n_tags=10, n_deps=5,
                               n_ent_types=2, seed=3)
    return [Example.from_doc(d) for d in docs]


def test_pretrain_loss_decreases_and_weights_roundtrip(tmp_path):
    torch.manual_seed(0)
    cfg = Config.from_disk(CFG)
    nlp = init_nlp(cfg, device="cpu", sample_size=16)
    examples = _corpus(nlp)
    losses = pretrain_tok2vec(nlp, lambda n: iter(examples), steps=80,
                              batch_docs=24, n_buckets=128,
                              learn_rate=5e-3, log_every=20,
                              log=lambda *a: None)
    assert len(losses) >= 3
    assert losses[-1] < losses[0], losses  # the masked objective is learnable
    path = save_tok2vec(nlp, tmp_path)
    ref = {k: v.clone() for k, v in nlp.tok2vec.module.state_dict().items()}
    # a FRESH pipeline gets the pretrained encoder via init_tok2vec
    torch.manual_seed(1)
    nlp2 = init_nlp(cfg, device="cpu", sample_size=16)
    before = nlp2.tok2vec.module.state_dict()
    assert any(not torch.allclose(before[k], ref[k]) for k in ref)
    n = load_init_tok2vec(nlp2, tmp_path)
    assert n == len(ref)
    after = nlp2.tok2vec.module.state_dict()
    for k in ref:
        assert torch.allclose(after[k], ref[k]), k
    # supervised training proceeds from the pretrained weights
    from spacy_ray_amd.config.config import resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine

    T = resolve(cfg.interpolate()["training"], validate=False)
    engine = ZeRO1Engine(nlp2, T["optimizer"], LocalComm())
    losses2 = {}
    engine.accumulate(examples[:24], drop=0.0, losses=losses2)
    engine.apply_step()
    assert all(np.isfinite(v) for v in losses2.values())
