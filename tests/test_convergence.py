"""Convergence smoke tests: each component learns a learnable synthetic
pattern well above chance within a small step budget (CPU)."""
import numpy as np
import pytest
import torch

from spacy_ray_amd.config.config import Config
from spacy_ray_amd.data.corpus import make_synthetic_docs
from spacy_ray_amd.pipeline.language import init_nlp
from spacy_ray_amd.train.scorer import score_examples
from spacy_ray_amd.train.stepper import SimpleStepper
from spacy_ray_amd.train.optimizer import make_adam
from spacy_ray_amd.vocab.doc import Example

PARSER_CFG = """
[nlp]
lang = "en"
pipeline = ["tok2vec", "parser", "ner"]

[components]

[components.tok2vec]
factory = "tok2vec"

[components.tok2vec.model]
@architectures = "spacy.HashEmbedCNN.v2"
width = 32
depth = 2
embed_size = 500

[components.parser]
factory = "parser"

[components.parser.model]
@architectures = "spacy.TransitionBasedParser.v2"
state_type = "parser"
hidden_width = 32

[components.parser.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 32

[components.ner]
factory = "ner"

[components.ner.model]
@architectures = "spacy.TransitionBasedParser.v2"
state_type = "ner"
hidden_width = 32

[components.ner.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 32

[corpora]

[corpora.train]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 100
words_per_doc = 8
vocab_size = 50
n_tags = 5
n_deps = 3
n_ent_types = 2
seed = 0

[corpora.dev]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 30
words_per_doc = 8
vocab_size = 50
n_tags = 5
n_deps = 3
n_ent_types = 2
seed = 1
shuffle = false

[training]
seed = 0
max_steps = 10
train_corpus = "corpora.train"
dev_corpus = "corpora.dev"

[training.batcher]
@batchers = "spacy.batch_by_words.v1"
size = 400

[training.optimizer]
@optimizers = "Adam.v1"
learn_rate = 0.003

[training.score_weights]
dep_uas = 1.0
"""


def test_parser_learns_chain_grammar():
    """Chain trees (head = previous token) are fully decodable from the
    transition system alone — the parser must reach high UAS fast."""
    cfg = Config.from_str(PARSER_CFG)
    nlp = init_nlp(cfg, sample_size=32)
    train_docs = make_synthetic_docs(
        nlp.vocab, n_docs=120, words_per_doc=8, vocab_size=50, n_tags=5,
        n_deps=3, n_ent_types=2, seed=0, tree_style="chain")
    dev_docs = make_synthetic_docs(
        nlp.vocab, n_docs=30, words_per_doc=8, vocab_size=50, n_tags=5,
        n_deps=3, n_ent_types=2, seed=1, tree_style="chain")
    train = [Example.from_doc(d) for d in train_docs]
    dev = [Example.from_doc(d) for d in dev_docs]
    stepper = SimpleStepper(nlp, make_adam(learn_rate=0.005))
    for i in range(30):
        batch = train[(i * 16) % 96:(i * 16) % 96 + 16]
        stepper.accumulate(batch, drop=0.0, losses={})
        stepper.apply_step()
    nlp.predict_docs([eg.predicted for eg in dev])
    scores = score_examples(dev, nlp.pipe_names)
    assert scores["dep_uas"] > 0.8, scores


def test_ner_learns_word_correlated_entities():
    cfg = Config.from_str(PARSER_CFG)
    nlp = init_nlp(cfg, sample_size=64)
    from spacy_ray_amd.config.config import resolve, resolve_dot_names

    icfg = cfg.interpolate()
    train_corpus, dev_corpus = resolve_dot_names(icfg, ["corpora.train", "corpora.dev"])
    train = list(train_corpus(nlp))
    dev = list(dev_corpus(nlp))
    stepper = SimpleStepper(nlp, make_adam(learn_rate=0.005))
    for i in range(40):
        batch = train[(i * 20) % 80:(i * 20) % 80 + 20]
        stepper.accumulate(batch, drop=0.0, losses={})
        stepper.apply_step()
    nlp.predict_docs([eg.predicted for eg in dev])
    scores = score_examples(dev, nlp.pipe_names)
    # entities are word-correlated spans; beating 0.2 F requires real signal
    # (untrained model emits ~0)
    assert scores["ents_f"] > 0.2, scores
