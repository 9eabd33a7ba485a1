"""End-to-end pipeline tests on CPU: build from config, train, converge,
checkpoint round-trip."""
import numpy as np
import pytest
import torch

from spacy_ray_amd.config.config import Config, resolve, resolve_dot_names
from spacy_ray_amd.pipeline.language import build_nlp, init_nlp
from spacy_ray_amd.train.loop import create_train_batches, train_while_improving
from spacy_ray_amd.train.stepper import SimpleStepper
from spacy_ray_amd.train.scorer import score_examples, weighted_score

TAGGER_CFG = """
[nlp]
lang = "en"
pipeline = ["tok2vec", "tagger"]

[components]

[components.tok2vec]
factory = "tok2vec"

[components.tok2vec.model]
@architectures = "spacy.HashEmbedCNN.v2"
width = 32
depth = 2
embed_size = 500

[components.tagger]
factory = "tagger"

[components.tagger.model]
@architectures = "spacy.Tagger.v2"

[components.tagger.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 32

[corpora]

[corpora.train]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 200
words_per_doc = 10
vocab_size = 100
n_tags = 5
seed = 0

[corpora.dev]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 50
words_per_doc = 10
vocab_size = 100
n_tags = 5
seed = 1
shuffle = false

[training]
seed = 0
dropout = 0.0
max_steps = 60
eval_frequency = 30
train_corpus = "corpora.train"
dev_corpus = "corpora.dev"

[training.batcher]
@batchers = "spacy.batch_by_words.v1"
size = 500

[training.optimizer]
@optimizers = "Adam.v1"
learn_rate = 0.005

[training.score_weights]
tag_acc = 1.0
"""


def _train(cfg_text, max_steps=60):
    cfg = Config.from_str(cfg_text)
    nlp = init_nlp(cfg)
    icfg = cfg.interpolate()
    T = resolve(icfg["training"], validate=False)
    train_corpus, dev_corpus = resolve_dot_names(icfg, [T["train_corpus"], T["dev_corpus"]])
    stepper = SimpleStepper(nlp, T["optimizer"])
    batches = create_train_batches(nlp, train_corpus, T["batcher"], 0)
    dev = list(dev_corpus(nlp))

    def evaluate():
        scores = nlp.evaluate(dev)
        return weighted_score(scores, icfg["training"]["score_weights"]), scores

    first_losses, last_losses = None, None
    for batch, info, is_best in train_while_improving(
        nlp, stepper, batches, evaluate=evaluate, dropout=0.0,
        max_steps=max_steps, eval_frequency=max_steps // 2,
    ):
        if info["step"] == 2:
            first_losses = dict(info["losses"])
        last_losses = info
    return nlp, first_losses, last_losses, evaluate


def test_tagger_learns():
    nlp, first, last, evaluate = _train(TAGGER_CFG)
    score, scores = evaluate()
    # deterministic word->tag mapping with 10% noise: accuracy must beat
    # majority-class baseline decisively after 60 steps
    assert scores["tag_acc"] > 0.5, scores


def test_deterministic_init():
    cfg = Config.from_str(TAGGER_CFG)
    nlp1 = init_nlp(cfg)
    nlp2 = init_nlp(Config.from_str(TAGGER_CFG))
    p1 = dict(nlp1.torch_module().named_parameters())
    p2 = dict(nlp2.torch_module().named_parameters())
    assert p1.keys() == p2.keys()
    for k in p1:
        assert torch.equal(p1[k], p2[k]), k


def test_checkpoint_roundtrip(tmp_path):
    nlp, _, _, _ = _train(TAGGER_CFG, max_steps=20)
    from spacy_ray_amd.data.corpus import make_synthetic_docs

    docs = make_synthetic_docs(nlp.vocab, n_docs=5, words_per_doc=8, vocab_size=100,
                               n_tags=5, n_deps=3, n_ent_types=2, seed=9)
    preds1 = [list(d.tags) for d in nlp.predict_docs([d.copy_unannotated() for d in docs])]
    nlp.to_disk(tmp_path / "model")
    cfg = Config.from_disk(tmp_path / "model" / "config.cfg")
    nlp2 = build_nlp(cfg)
    nlp2.from_disk(tmp_path / "model")
    preds2 = [list(d.tags) for d in nlp2.predict_docs([d.copy_unannotated() for d in docs])]
    assert preds1 == preds2


def test_parser_ner_train_step():
    cfg = Config.from_disk("examples/configs/en_core_cnn.cfg")
    nlp = init_nlp(cfg, sample_size=32)
    icfg = cfg.interpolate()
    T = resolve(icfg["training"], validate=False)
    train_corpus, _ = resolve_dot_names(icfg, [T["train_corpus"], T["dev_corpus"]])
    examples = []
    for eg in train_corpus(nlp):
        examples.append(eg)
        if len(examples) >= 16:
            break
    stepper = SimpleStepper(nlp, T["optimizer"])
    losses = {}
    stepper.accumulate(examples, drop=0.0, losses=losses)
    stepper.apply_step()
    assert set(losses) == {"tagger", "parser", "ner"}
    assert all(np.isfinite(v) and v > 0 for v in losses.values()), losses


def test_nlp_pipe_and_call():
    cfg = Config.from_str(TAGGER_CFG)
    nlp = init_nlp(cfg, sample_size=16)
    docs = list(nlp.pipe(["w1 w2 w3", "w4 w5"], batch_size=2))
    assert len(docs) == 2
    assert docs[0].tags is not None and len(docs[0].tags) == 3
    d = nlp("w1 w2")
    assert d.tags is not None and len(d) == 2


def test_xx_multilingual_config_trains_cpu():
    cfg = Config.from_disk("examples/configs/xx_multilingual.cfg")
    nlp = init_nlp(cfg, sample_size=24)
    assert nlp.pipe_names == ["tok2vec", "tagger", "parser"]
    icfg = cfg.interpolate()
    T = resolve(icfg["training"], validate=False)
    train_corpus, _ = resolve_dot_names(icfg, [T["train_corpus"], T["dev_corpus"]])
    examples = []
    for eg in train_corpus(nlp):
        examples.append(eg)
        if len(examples) >= 12:
            break
    stepper = SimpleStepper(nlp, T["optimizer"])
    losses = {}
    stepper.accumulate(examples, drop=0.0, losses=losses)
    stepper.apply_step()
    assert losses["tagger"] > 0 and losses["parser"] > 0


def test_predict_sets_annotations():
    cfg = Config.from_disk("examples/configs/en_core_cnn.cfg")
    nlp = init_nlp(cfg, sample_size=16)
    from spacy_ray_amd.data.corpus import make_synthetic_docs

    docs = make_synthetic_docs(nlp.vocab, n_docs=3, words_per_doc=10, vocab_size=100,
                               n_tags=5, n_deps=3, n_ent_types=2, seed=3)
    outs = nlp.predict_docs([d.copy_unannotated() for d in docs])
    for d in outs:
        assert d.tags is not None and len(d.tags) == len(d)
        assert d.heads is not None and len(d.heads) == len(d)
        assert d.deps is not None
        assert d.ents is not None
        # heads are in-range or -1 (root)
        assert all(-1 <= int(h) < len(d) for h in d.heads)


EMBEDDED_TAGGER_CFG = """
[nlp]
lang = "en"
pipeline = ["tagger"]

[components]

[components.tagger]
factory = "tagger"

[components.tagger.model]
@architectures = "spacy.Tagger.v2"

[components.tagger.model.tok2vec]
@architectures = "spacy.HashEmbedCNN.v2"
width = 32
depth = 2
embed_size = 500

[corpora]

[corpora.train]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 200
words_per_doc = 10
vocab_size = 100
n_tags = 5
seed = 0

[corpora.dev]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 50
words_per_doc = 10
vocab_size = 100
n_tags = 5
seed = 1
shuffle = false

[training]
train_corpus = "corpora.train"
dev_corpus = "corpora.dev"
max_steps = 40
eval_frequency = 20

[training.optimizer]
@optimizers = "Adam.v1"
learn_rate = 0.01
"""


def test_embedded_tok2vec_tagger(tmp_path):
    """A pipe with a FULL tok2vec block (not a listener) owns its encoder:
    no shared tok2vec pipe in the pipeline, params live inside the pipe's
    module, training learns, checkpoints roundtrip."""
    import torch

    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.pipeline.language import build_nlp, init_nlp

    cfg = Config.from_str(EMBEDDED_TAGGER_CFG)
    nlp = init_nlp(cfg)
    assert nlp.pipe_names == ["tagger"]
    tagger = dict(nlp.pipeline)["tagger"]
    assert hasattr(tagger.module, "embedded_t2v")
    n_emb = sum(p.numel() for p in tagger.module.embedded_t2v.parameters())
    assert n_emb > 0

    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.config.config import resolve

    T = resolve(cfg.interpolate()["training"], validate=False)
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    # the embedded encoder's params are in the flat buffer
    assert any("embedded_t2v" in n for n in engine.param_names)
    from spacy_ray_amd.config.config import resolve_dot_names
    (train_corpus,) = resolve_dot_names(cfg.interpolate(), ["corpora.train"])
    examples = list(train_corpus(nlp))[:64]
    first = last = None
    for i in range(20):
        losses = {}
        engine.accumulate(examples, losses=losses)
        engine.apply_step()
        if first is None:
            first = losses["tagger"]
        last = losses["tagger"]
    assert last < first, (first, last)

    # predict + roundtrip
    doc = nlp("hello world")
    assert doc.tags and len(doc.tags) == 2
    nlp.to_disk(tmp_path / "m")
    nlp2 = build_nlp(Config.from_disk(tmp_path / "m" / "config.cfg"))
    nlp2.from_disk(tmp_path / "m")
    t2 = dict(nlp2.pipeline)["tagger"]
    a = torch.cat([p.reshape(-1) for p in tagger.module.embedded_t2v.parameters()])
    b = torch.cat([p.reshape(-1) for p in t2.module.embedded_t2v.parameters()])
    assert torch.allclose(a.float(), b.float())


def test_checkpoint_roundtrip_full_pipeline(tmp_path):
    """en_core (tok2vec+tagger+parser+NER): saved and reloaded pipelines must
    produce IDENTICAL annotations (tags, heads, deps, BILUO ents) — covers
    the parser/NER label-state + transition-model checkpoint path."""
    from spacy_ray_amd.data.corpus import make_synthetic_docs

    cfg = Config.from_disk("examples/configs/en_core_cnn.cfg")
    nlp = init_nlp(cfg, sample_size=32)
    docs = make_synthetic_docs(nlp.vocab, n_docs=6, words_per_doc=9,
                               vocab_size=150, n_tags=50, n_deps=40,
                               n_ent_types=4, seed=21)

    def annotate(model):
        outs = model.predict_docs([d.copy_unannotated() for d in docs])
        return [(list(d.tags), list(d.heads), list(d.deps), list(d.ents))
                for d in outs]

    preds1 = annotate(nlp)
    nlp.to_disk(tmp_path / "model")
    nlp2 = build_nlp(Config.from_disk(tmp_path / "model" / "config.cfg"))
    nlp2.from_disk(tmp_path / "model")
    assert annotate(nlp2) == preds1


def test_listener_width_mismatch_fails_at_build():
    cfg_text = TAGGER_CFG.replace(
        '@architectures = "spacy.Tok2VecListener.v1"\nwidth = 32',
        '@architectures = "spacy.Tok2VecListener.v1"\nwidth = 64')
    assert "width = 64" in cfg_text
    with pytest.raises(ValueError, match="width"):
        build_nlp(Config.from_str(cfg_text))


def test_listener_without_shared_tok2vec_fails_at_build():
    import re

    cfg_text = TAGGER_CFG.replace('pipeline = ["tok2vec", "tagger"]',
                                  'pipeline = ["tagger"]')
    with pytest.raises(ValueError, match="listens to a shared tok2vec"):
        build_nlp(Config.from_str(cfg_text))


def test_nlp_pipe_streaming():
    """nlp.pipe: streams texts/Docs in batches, yields annotated Docs in
    order (spaCy nlp.pipe contract)."""
    from spacy_ray_amd.vocab.doc import Doc

    nlp = init_nlp(Config.from_str(TAGGER_CFG), sample_size=16)
    texts = [f"word{i} and word{i + 1}" for i in range(7)]
    inputs = list(texts[:5]) + [Doc(nlp.vocab, ["premade", "doc"])] + [texts[6]]
    outs = list(nlp.pipe(inputs, batch_size=3))
    assert len(outs) == 7
    assert outs[5].words == ["premade", "doc"]
    for d in outs:
        assert d.tags is not None and len(d.tags) == len(d)


def test_annotating_components_set_predictions_before_loss():
    """training.annotating_components: listed pipes write their predictions
    onto eg.predicted before losses run (spaCy contract), so downstream
    user components can consume them."""
    from spacy_ray_amd.config.config import resolve_dot_names

    cfg = Config.from_str(TAGGER_CFG)
    nlp = init_nlp(cfg, sample_size=16)
    icfg = cfg.interpolate()
    (train_corpus,) = resolve_dot_names(icfg, ["corpora.train"])
    examples = []
    for eg in train_corpus(nlp):
        examples.append(eg)
        if len(examples) >= 4:
            break
    for eg in examples:
        eg.predicted.tags = None
    nlp._annotating = ["tagger"]
    try:
        total, losses = nlp.forward_loss(examples)
    finally:
        nlp._annotating = []
    assert "tagger" in losses
    for eg in examples:
        assert eg.predicted.tags is not None
        assert len(eg.predicted.tags) == len(eg.predicted)
    total.backward()  # annotation ran under no_grad; loss graph intact


@pytest.mark.parametrize("pipes", [["tok2vec", "parser"], ["tok2vec", "ner"]])
def test_single_head_pipelines_train_and_predict(pipes):
    """Unusual pipeline shapes (parser-only / NER-only heads) must train and
    annotate — no hidden dependency on the full en_core lineup."""
    cfg = Config.from_disk("examples/configs/en_core_cnn.cfg")
    cfg["nlp"]["pipeline"] = list(pipes)
    for extra in ("tagger", "parser", "ner"):
        if extra not in pipes and extra in cfg["components"]:
            del cfg["components"][extra]
    nlp = init_nlp(cfg, sample_size=24)
    assert nlp.pipe_names == pipes
    icfg = cfg.interpolate()
    T = resolve(icfg["training"], validate=False)
    (train_corpus,) = resolve_dot_names(icfg, ["corpora.train"])
    examples = []
    for eg in train_corpus(nlp):
        examples.append(eg)
        if len(examples) >= 12:
            break
    stepper = SimpleStepper(nlp, T["optimizer"])
    losses = {}
    stepper.accumulate(examples, losses=losses)
    stepper.apply_step()
    head = pipes[1]
    assert head in losses and np.isfinite(losses[head])
    docs = nlp.predict_docs([eg.predicted.copy_unannotated() for eg in examples[:3]])
    for d in docs:
        if head == "parser":
            assert d.heads is not None and len(d.heads) == len(d)
        else:
            assert d.ents is not None and len(d.ents) == len(d)


def test_annotating_components_transition_pipe():
    """annotating_components with a transition pipe (parser): predictions
    land on eg.predicted through the shard-split decode path."""
    cfg = Config.from_disk("examples/configs/en_core_cnn.cfg")
    nlp = init_nlp(cfg, sample_size=24)
    icfg = cfg.interpolate()
    (train_corpus,) = resolve_dot_names(icfg, ["corpora.train"])
    examples = []
    for eg in train_corpus(nlp):
        examples.append(eg)
        if len(examples) >= 6:
            break
    for eg in examples:
        eg.predicted.heads = None
    nlp._annotating = ["parser", "ner"]
    try:
        total, losses = nlp.forward_loss(examples)
    finally:
        nlp._annotating = []
    for eg in examples:
        assert eg.predicted.heads is not None
        assert len(eg.predicted.heads) == len(eg.predicted)
        assert eg.predicted.ents is not None
    total.backward()


def test_patience_is_step_based():
    """spaCy's patience is in STEPS: with eval_frequency=2 and
    patience=4, a never-improving run stops ~4 steps after the first eval."""
    from spacy_ray_amd.vocab.doc import Doc, Example, Vocab

    cfg = Config.from_str(TAGGER_CFG)
    nlp = init_nlp(cfg)
    icfg = cfg.interpolate()
    T = resolve(icfg["training"], validate=False)
    train_corpus = resolve_dot_names(icfg, [T["train_corpus"]])[0]
    stepper = SimpleStepper(nlp, T["optimizer"])
    batches = create_train_batches(nlp, train_corpus, T["batcher"], 0)
    calls = {"n": 0}

    def evaluate():  # strictly decreasing: never improves after the first
        calls["n"] += 1
        return 1.0 - 0.01 * calls["n"], {}

    steps = [
        info["step"]
        for _, info, _ in train_while_improving(
            nlp, stepper, batches, evaluate=evaluate, dropout=0.0,
            max_steps=100, eval_frequency=2, patience=4,
        )
    ]
    # first eval at step 2 (best); stop once step - best_step >= 4 -> step 6
    assert steps[-1] <= 8, steps


def test_label_discovery_sees_late_labels():
    """An entity type first appearing after the 128-example init sample must
    not crash mid-training: labels are discovered over the FULL
    corpus."""
    from spacy_ray_amd.config.registry import registry
    from spacy_ray_amd.vocab.doc import Doc, Example

    @registry.readers("test.LateLabelCorpus.v1")
    def late_label_corpus():
        def corpus(nlp):
            for i in range(200):
                ents = ["U-LATE" if i >= 150 else "O", "O"]
                yield Example.from_doc(
                    Doc(nlp.vocab, [f"w{i % 7}", "x"], tags=["A", "B"],
                        ents=ents))
        return corpus

    cfg_text = """
[nlp]
lang = "en"
pipeline = ["tok2vec", "ner"]

[components]

[components.tok2vec]
factory = "tok2vec"

[components.tok2vec.model]
@architectures = "spacy.HashEmbedCNN.v2"
width = 32
depth = 1
embed_size = 100

[components.ner]
factory = "ner"

[components.ner.model]
@architectures = "spacy.TransitionBasedParser.v2"
state_type = "ner"

[components.ner.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 32

[corpora]

[corpora.train]
@readers = "test.LateLabelCorpus.v1"

[training]
seed = 0
train_corpus = "corpora.train"
dev_corpus = "corpora.train"
"""
    nlp = init_nlp(Config.from_str(cfg_text))
    ner = nlp.get_pipe("ner")
    assert "LATE" in ner.labels
    # and training on a late batch works (would KeyError before the fix)
    icfg = Config.from_str(cfg_text).interpolate()
    train_corpus = resolve_dot_names(icfg, ["corpora.train"])[0]
    egs = list(train_corpus(nlp))[150:160]
    nlp.forward_loss(egs)


def test_explicit_labels_in_component_config():
    cfg_text = TAGGER_CFG.replace(
        '[components.tagger]\nfactory = "tagger"',
        '[components.tagger]\nfactory = "tagger"\nlabels = ["TAG0","TAG1","TAG2","TAG3","TAG4"]',
    )
    nlp = init_nlp(Config.from_str(cfg_text))
    assert nlp.get_pipe("tagger").labels == ["TAG0", "TAG1", "TAG2", "TAG3", "TAG4"]


def test_select_pipes_disables_inference():
    from spacy_ray_amd.vocab.doc import Doc

    cfg = Config.from_str(TAGGER_CFG)
    nlp = init_nlp(cfg)
    doc = Doc(nlp.vocab, ["a", "b"])
    with nlp.select_pipes(disable=["tagger"]):
        nlp.predict_docs([doc])
        assert doc.tags is None  # disabled pipe did NOT annotate
    nlp.predict_docs([doc])
    assert doc.tags is not None  # re-enabled


def test_missing_biluo_excluded_from_loss():
    """'-' gold tokens produce all-zero gold rows (no supervision) instead of
    negative O supervision."""
    from spacy_ray_amd import _srx_cpu
    from spacy_ray_amd.vocab.doc import biluo_to_codes
    import numpy as np

    codes = biluo_to_codes(["U-PER", "-", "O"], {"PER": 0})
    assert codes.tolist() == [4, -1, 0]
    batch = _srx_cpu.BiluoBatch(np.array([3], dtype=np.int32), 1, 0)
    batch.set_gold(codes.astype(np.int32))
    # step to token 1 (the missing one): advance with UNIT-PER then inspect
    act, feats, valid, gold = batch.step_arrays(True)
    batch.advance(np.array([4], dtype=np.int32))  # U-PER on token 0
    act, feats, valid, gold = batch.step_arrays(True)
    assert gold[0].sum() == 0  # all-zero gold row -> masked out of the CE


def test_parser_use_break_learns_sentence_boundaries():
    """use_break=true: the parser trains through BREAK transitions and
    annotates sent_starts at predict time."""
    import numpy as np

    from spacy_ray_amd.config.registry import registry
    from spacy_ray_amd.vocab.doc import Doc, Example

    @registry.readers("test.SentCorpus.v1")
    def sent_corpus():
        def corpus(nlp):
            rng = np.random.RandomState(0)
            for i in range(300):
                # two 3-token "sentences": w-root chain per sentence
                words = [f"a{rng.randint(6)}", "b", "c",
                         f"d{rng.randint(6)}", "e", "f"]
                heads = [-1, 0, 0, -1, 3, 3]
                deps = ["ROOT", "x", "y", "ROOT", "x", "y"]
                sents = [1, 0, 0, 1, 0, 0]
                yield Example.from_doc(Doc(nlp.vocab, words, heads=heads,
                                           deps=deps, sent_starts=sents))
        return corpus

    cfg_text = """
[nlp]
lang = "en"
pipeline = ["tok2vec", "parser"]

[components]

[components.tok2vec]
factory = "tok2vec"

[components.tok2vec.model]
@architectures = "spacy.HashEmbedCNN.v2"
width = 32
depth = 1
embed_size = 200

[components.parser]
factory = "parser"
use_break = true

[components.parser.model]
@architectures = "spacy.TransitionBasedParser.v2"
state_type = "parser"

[components.parser.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 32

[corpora]

[corpora.train]
@readers = "test.SentCorpus.v1"

[training]
seed = 0
train_corpus = "corpora.train"
dev_corpus = "corpora.train"

[training.optimizer]
@optimizers = "Adam.v1"
learn_rate = 0.01
"""
    cfg = Config.from_str(cfg_text)
    nlp = init_nlp(cfg)
    parser = nlp.get_pipe("parser")
    assert parser.use_break
    assert parser._n_actions() == 2 + 2 * len(parser.labels) + 1
    icfg = cfg.interpolate()
    train_corpus = resolve_dot_names(icfg, ["corpora.train"])[0]
    T = resolve(icfg["training"], validate=False)
    stepper = SimpleStepper(nlp, T["optimizer"])
    egs = list(train_corpus(nlp))
    for _ in range(30):
        stepper.accumulate(egs[:64], drop=0.0, losses={})
        stepper.apply_step()
    docs = [eg.predicted for eg in egs[:8]]
    nlp.predict_docs(docs)
    hits = total = 0
    for d in docs:
        assert d.sent_starts is not None
        assert d.sent_starts[0] == 1
        hits += int(d.sent_starts[3] == 1)
        total += 1
    assert hits >= total * 0.7, (hits, total)  # boundary learned


TEXTCAT_CFG = """
[nlp]
lang = "en"
pipeline = ["tok2vec", "textcat", "senter"]

[components]

[components.tok2vec]
factory = "tok2vec"

[components.tok2vec.model]
@architectures = "spacy.HashEmbedCNN.v2"
width = 64
depth = 2
embed_size = 500
window_size = 1
maxout_pieces = 3
subword_features = true
pretrained_vectors = null

[components.textcat]
factory = "textcat"

[components.textcat.model]
@architectures = "spacy.TextCatReduce.v1"
exclusive_classes = true

[components.textcat.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64

[components.senter]
factory = "senter"

[components.senter.model]
@architectures = "spacy.Tagger.v2"

[components.senter.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64

[corpora]

[corpora.train]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 96
words_per_doc = 14
vocab_size = 120
n_tags = 10
seed = 5

[corpora.dev]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 32
seed = 6

[training]
seed = 0
train_corpus = "corpora.train"
dev_corpus = "corpora.dev"

[training.optimizer]
@optimizers = "Adam.v1"
learn_rate = 0.01
"""


def test_textcat_and_senter_train_and_predict():
    """Doc-classification (textcat) + sentence recognizer (senter) pipes:
    losses fall, predictions populate Doc.cats / Doc.sent_starts, the
    scorer reports cats_macro_acc and sents_f, and the learnable synthetic
    signals are actually learned above chance."""
    import torch

    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.train.scorer import score_examples
    from spacy_ray_amd.vocab.doc import Example

    torch.manual_seed(0)
    cfg = Config.from_str(TEXTCAT_CFG)
    nlp = init_nlp(cfg, device="cpu", sample_size=64)
    docs = make_synthetic_docs(nlp.vocab, n_docs=96, words_per_doc=14,
                               vocab_size=120, n_tags=10, n_deps=10,
                               n_ent_types=3, seed=5)
    assert docs[0].cats and abs(sum(docs[0].cats.values()) - 1.0) < 1e-6
    assert docs[0].sent_starts is not None
    examples = [Example.from_doc(d) for d in docs]
    opt = {"@optimizers": "Adam.v1", "learn_rate": 0.01}
    from spacy_ray_amd.config.config import resolve

    engine = ZeRO1Engine(nlp, resolve({"o": opt}, validate=False)["o"], LocalComm())
    first = last = None
    for i in range(30):
        losses = {}
        engine.accumulate(examples, drop=0.0, losses=losses)
        engine.apply_step()
        if i == 0:
            first = dict(losses)
        last = dict(losses)
    assert last["textcat"] < first["textcat"]
    assert last["senter"] < first["senter"]
    nlp.predict_docs([eg.predicted for eg in examples])
    scores = score_examples(examples, ["textcat", "senter"])
    assert scores["cats_macro_acc"] > 0.5, scores
    assert scores["sents_f"] > 0.6, scores
    for eg in examples[:3]:
        assert eg.predicted.cats and len(eg.predicted.cats) >= 3
        assert eg.predicted.sent_starts is not None


def test_textcat_senter_checkpoint_roundtrip(tmp_path):
    """to_disk/load for the textcat+senter pipeline: predictions identical
    after the roundtrip (generic component serialization covers new pipes)."""
    import numpy as np

    import spacy_ray_amd
    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.pipeline.language import init_nlp

    cfg = Config.from_disk("examples/configs/en_textcat.cfg")
    nlp = init_nlp(cfg, device="cpu", sample_size=32)
    docs = make_synthetic_docs(nlp.vocab, n_docs=8, words_per_doc=12,
                               vocab_size=120, n_tags=10, n_deps=10,
                               n_ent_types=3, seed=5)
    nlp.to_disk(str(tmp_path))
    nlp2 = spacy_ray_amd.load(str(tmp_path), device="cpu")
    outs1 = nlp.predict_docs([d.copy_unannotated() for d in docs])
    outs2 = nlp2.predict_docs([d.copy_unannotated() for d in docs])
    for o1, o2 in zip(outs1, outs2):
        for k in o1.cats:
            assert abs(o1.cats[k] - o2.cats[k]) < 1e-5
        assert (o1.sent_starts == o2.sent_starts).all()


def test_entity_ruler_patterns_and_pipeline():
    """entity_ruler: phrase + token-spec patterns, longest-match,
    existing-entity preservation, checkpoint roundtrip via cfg.json."""
    from spacy_ray_amd.pipeline.ruler import EntityRulerPipe
    from spacy_ray_amd.vocab.doc import Doc, Vocab

    vocab = Vocab()
    ruler = EntityRulerPipe("entity_ruler")
    ruler.add_patterns([
        {"label": "ORG", "pattern": "Acme Corp"},
        {"label": "GPE", "pattern": [{"LOWER": "san"}, {"LOWER": "francisco"}]},
        {"label": "CARDINAL", "pattern": [{"IS_DIGIT": True}]},
        {"label": "ORG", "pattern": "Acme"},  # shorter: longest must win
    ])
    doc = Doc(vocab, ["Acme", "Corp", "opened", "in", "San", "Francisco",
                      "in", "1999"])
    ruler([doc])
    assert doc.ents == ["B-ORG", "L-ORG", "O", "O", "B-GPE", "L-GPE", "O",
                        "U-CARDINAL"]
    # existing entities preserved by default
    doc2 = Doc(vocab, ["Acme", "rocks"], ents=["U-PRODUCT", "O"])
    ruler([doc2])
    assert doc2.ents == ["U-PRODUCT", "O"]
    # overwrite_ents=True replaces
    ruler2 = EntityRulerPipe("entity_ruler", overwrite_ents=True,
                             patterns=[{"label": "ORG", "pattern": "Acme"}])
    ruler2([doc2])
    assert doc2.ents == ["U-ORG", "O"]
    # cfg roundtrip
    cfg = ruler.state_cfg()
    r3 = EntityRulerPipe("entity_ruler")
    r3.load_cfg(cfg, "cpu")
    doc3 = Doc(vocab, ["San", "Francisco"])
    r3([doc3])
    assert doc3.ents == ["B-GPE", "L-GPE"]
    # unsupported keys fail loudly
    import pytest as _pytest

    with _pytest.raises(ValueError):
        ruler.add_patterns([{"label": "X", "pattern": [{"REGEX": "a+"}]}])


def test_entity_ruler_in_language_pipeline():
    """entity_ruler inside a config-built pipeline: runs in predict,
    skipped in training, survives to_disk/load."""
    import spacy_ray_amd
    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.pipeline.language import init_nlp

    cfg_text = TEXTCAT_CFG.replace(
        'pipeline = ["tok2vec", "textcat", "senter"]',
        'pipeline = ["tok2vec", "textcat", "senter", "entity_ruler"]',
    ).replace(
        "[corpora]",
        """[components.entity_ruler]
factory = "entity_ruler"

[corpora]""",
    )
    cfg = Config.from_str(cfg_text)
    nlp = init_nlp(cfg, device="cpu", sample_size=16)
    nlp.get_pipe("entity_ruler").add_patterns(
        [{"label": "WORD", "pattern": "w1"}])
    from spacy_ray_amd.data.corpus import make_synthetic_docs

    docs = make_synthetic_docs(nlp.vocab, n_docs=4, words_per_doc=10,
                               vocab_size=30, n_tags=5, n_deps=5,
                               n_ent_types=2, seed=2)
    outs = nlp.predict_docs([d.copy_unannotated() for d in docs])
    hits = sum(e == "U-WORD" for d in outs for e in (d.ents or []))
    assert hits > 0  # w1 is frequent under the Zipf lexicon
    import tempfile

    with tempfile.TemporaryDirectory() as td:
        nlp.to_disk(td)
        nlp2 = spacy_ray_amd.load(td, device="cpu")
        outs2 = nlp2.predict_docs([d.copy_unannotated() for d in docs])
        assert [d.ents for d in outs2] == [d.ents for d in outs]


def test_morphologizer_trains_and_predicts():
    """morphologizer over CoNLL-U-style FEATS strings: learns a
    word-deterministic morph signal, predicts into Doc.morphs, scores as
    morph_acc; missing FEATS ('' after convert) are excluded from loss."""
    import numpy as np
    import torch

    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.train.scorer import score_examples
    from spacy_ray_amd.vocab.doc import Example

    cfg_text = TEXTCAT_CFG.replace(
        'pipeline = ["tok2vec", "textcat", "senter"]',
        'pipeline = ["tok2vec", "morphologizer"]',
    ).replace("""[components.textcat]
factory = "textcat"

[components.textcat.model]
@architectures = "spacy.TextCatReduce.v1"
exclusive_classes = true

[components.textcat.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64

[components.senter]
factory = "senter"

[components.senter.model]
@architectures = "spacy.Tagger.v2"

[components.senter.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64""",
"""[components.morphologizer]
factory = "morphologizer"

[components.morphologizer.model]
@architectures = "spacy.Tagger.v2"

[components.morphologizer.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64""")
    torch.manual_seed(0)
    cfg = Config.from_str(cfg_text)
    from spacy_ray_amd.data.corpus import make_synthetic_docs

    # init_nlp discovers labels from the corpus sample: attach morphs to a
    # synthetic world where morph = f(tag) (fully learnable)
    nlp = init_nlp(cfg, device="cpu", sample_size=8)
    docs = make_synthetic_docs(nlp.vocab, n_docs=64, words_per_doc=12,
                               vocab_size=100, n_tags=8, n_deps=5,
                               n_ent_types=2, seed=3)
    for d in docs:
        d.morphs = [f"Feat={t}" for t in d.tags]
    # re-initialize the pipe with morph-bearing examples for label discovery
    pipe = nlp.get_pipe("morphologizer")
    pipe.labels = []
    pipe.module = None  # rebuild the head for the discovered label count
    examples = [Example.from_doc(d) for d in docs]
    pipe.initialize(examples, "cpu")
    opt = {"@optimizers": "Adam.v1", "learn_rate": 0.01}
    engine = ZeRO1Engine(nlp, resolve({"o": opt}, validate=False)["o"],
                         LocalComm())
    first = last = None
    for i in range(25):
        losses = {}
        engine.accumulate(examples, drop=0.0, losses=losses)
        engine.apply_step()
        if i == 0:
            first = dict(losses)
        last = dict(losses)
    assert last["morphologizer"] < first["morphologizer"]
    nlp.predict_docs([eg.predicted for eg in examples])
    scores = score_examples(examples, ["morphologizer"])
    assert scores["morph_acc"] > 0.5, scores


def test_spancat_trains_and_predicts():
    """spancat: ngram suggester + multilabel span classifier over prefix-sum
    pooling; learns word-deterministic (possibly overlapping) spans and
    writes Doc.spans[spans_key]; scored as spans_sc_f."""
    import numpy as np
    import torch

    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.train.scorer import score_examples
    from spacy_ray_amd.vocab.doc import Example

    cfg_text = TEXTCAT_CFG.replace(
        'pipeline = ["tok2vec", "textcat", "senter"]',
        'pipeline = ["tok2vec", "spancat"]',
    ).replace("""[components.textcat]
factory = "textcat"

[components.textcat.model]
@architectures = "spacy.TextCatReduce.v1"
exclusive_classes = true

[components.textcat.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64

[components.senter]
factory = "senter"

[components.senter.model]
@architectures = "spacy.Tagger.v2"

[components.senter.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64""",
"""[components.spancat]
factory = "spancat"
spans_key = "sc"
max_ngram = 2

[components.spancat.model]
@architectures = "spacy.SpanCategorizer.v1"

[components.spancat.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64""")
    torch.manual_seed(0)
    cfg = Config.from_str(cfg_text)
    nlp = init_nlp(cfg, device="cpu", sample_size=8)
    docs = make_synthetic_docs(nlp.vocab, n_docs=64, words_per_doc=10,
                               vocab_size=60, n_tags=6, n_deps=5,
                               n_ent_types=2, seed=7)
    # word-deterministic spans: every token whose id ends in 0 is a LONE
    # span; every bigram starting at an even word id is a PAIR span
    for d in docs:
        spans = []
        for i, w in enumerate(d.words):
            wid = int(w[1:])
            if wid % 10 == 0:
                spans.append((i, i + 1, "LONE"))
            if wid % 2 == 0 and i + 2 <= len(d):
                spans.append((i, i + 2, "PAIR"))
        d.spans["sc"] = spans
    pipe = nlp.get_pipe("spancat")
    pipe.labels = []
    pipe.module = None  # rebuild the head for the discovered label count
    examples = [Example.from_doc(d) for d in docs]
    pipe.initialize(examples, "cpu")
    assert pipe.labels == ["LONE", "PAIR"]
    opt = {"@optimizers": "Adam.v1", "learn_rate": 0.02}
    engine = ZeRO1Engine(nlp, resolve({"o": opt}, validate=False)["o"],
                         LocalComm())
    first = last = None
    for i in range(40):
        losses = {}
        engine.accumulate(examples, drop=0.0, losses=losses)
        engine.apply_step()
        if i == 0:
            first = dict(losses)
        last = dict(losses)
    assert last["spancat"] < first["spancat"]
    nlp.predict_docs([eg.predicted for eg in examples])
    scores = score_examples(examples, ["spancat"])
    assert scores["spans_sc_f"] > 0.5, scores
    # overlapping spans survive (the point of spancat vs NER)
    any_overlap = any(
        s1 != s2 and s1[0] < s2[1] and s2[0] < s1[1]
        for eg in examples for s1 in eg.predicted.spans.get("sc", [])
        for s2 in eg.predicted.spans.get("sc", [])
    )
    assert any_overlap


def test_beam_decode_parser_and_ner():
    """Beam decoding (decode-only, spaCy's beam parser/NER role): width-1
    beam equals the greedy decoder exactly (same model, same tie-breaks up
    to float order), larger widths return valid structures with total
    log-prob >= the width-1 path's."""
    import numpy as np
    import torch

    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.pipeline.beam import beam_decode, _ArcEagerState
    from spacy_ray_amd.pipeline.language import init_nlp

    torch.manual_seed(0)
    cfg = Config.from_disk("examples/configs/en_core_cnn.cfg")
    nlp = init_nlp(cfg, device="cpu", sample_size=24)
    docs = make_synthetic_docs(nlp.vocab, n_docs=6, words_per_doc=9,
                               vocab_size=80, n_tags=50, n_deps=40,
                               n_ent_types=4, seed=4)
    greedy = nlp.predict_docs([d.copy_unannotated() for d in docs])
    parser = nlp.get_pipe("parser")
    ner = nlp.get_pipe("ner")
    from spacy_ray_amd.models.batch import TokenBatch

    for width in (1, 4):
        parser.beam_width = width
        ner.beam_width = width
        outs = nlp.predict_docs([d.copy_unannotated() for d in docs])
        for d in outs:
            assert len(d.heads) == len(d)
            assert all(-1 <= int(h) < len(d) for h in d.heads)
            assert len(d.ents) == len(d)
        if width == 1:
            for g, b in zip(greedy, outs):
                assert (np.asarray(g.heads) == np.asarray(b.heads)).all()
                assert g.ents == b.ents
    # beam total log-prob is monotone in width (same scoring model)
    batch = TokenBatch([d.copy_unannotated() for d in docs],
                       torch.device("cpu"))
    with torch.no_grad():
        t2v = nlp.tok2vec.forward(batch)
    s1 = beam_decode(parser, docs, t2v, 1, _ArcEagerState)
    s4 = beam_decode(parser, docs, t2v, 4, _ArcEagerState)
    for a, b in zip(s1, s4):
        assert b.score >= a.score - 1e-4
    parser.beam_width = 1
    ner.beam_width = 1


def test_lemmatizer_and_attribute_ruler():
    """Rule lemmatizer (built-in English rules + irregulars, lookup mode)
    and attribute_ruler (pattern-driven TAG/LEMMA/MORPH overrides) — the
    two rule components completing the en_core_web_sm lineup."""
    from spacy_ray_amd.pipeline.attr_ruler import AttributeRulerPipe
    from spacy_ray_amd.pipeline.lemmatizer import LemmatizerPipe
    from spacy_ray_amd.vocab.doc import Doc, Vocab

    vocab = Vocab()
    lem = LemmatizerPipe("lemmatizer")
    doc = Doc(vocab, ["The", "children", "were", "running", "and",
                      "stopped", "near", "better", "cities"],
              tags=["DT", "NNS", "VBD", "VBG", "CC", "VBD", "IN", "JJR",
                    "NNS"])
    lem([doc])
    assert doc.lemmas == ["the", "child", "be", "run", "and", "stop",
                          "near", "good", "city"]
    # lookup mode
    lem2 = LemmatizerPipe("lemmatizer", mode="lookup",
                          lookups={"went": "go"})
    doc2 = Doc(vocab, ["went", "home"])
    lem2([doc2])
    assert doc2.lemmas == ["go", "home"]
    # attribute ruler overrides the tagger + feeds the lemmatizer
    ar = AttributeRulerPipe("attribute_ruler")
    ar.add_patterns([
        {"patterns": [[{"LOWER": "wo"}, {"LOWER": "n't"}]],
         "attrs": {"LEMMA": "will"}, "index": 0},
        {"patterns": [[{"LOWER": "wo"}, {"LOWER": "n't"}]],
         "attrs": {"LEMMA": "not", "TAG": "RB"}, "index": 1},
    ])
    doc3 = Doc(vocab, ["I", "wo", "n't", "go"],
               tags=["PRP", "MD", "RB", "VB"])
    ar([doc3])
    lem([doc3])
    assert doc3.lemmas[1] == "will" and doc3.lemmas[2] == "not"
    assert doc3.tags[2] == "RB"
    # roundtrip through cfg
    cfg = ar.state_cfg()
    ar2 = AttributeRulerPipe("attribute_ruler")
    ar2.load_cfg(cfg, "cpu")
    doc4 = Doc(vocab, ["wo", "n't"], tags=["MD", "RB"])
    ar2([doc4])
    assert doc4.lemmas[0] == "will"
    import pytest as _pytest

    with _pytest.raises(ValueError):
        ar.add_patterns([{"patterns": [[{"LOWER": "x"}]],
                          "attrs": {"DEP": "x"}}])
    # scorer integration
    from spacy_ray_amd.train.scorer import score_examples
    from spacy_ray_amd.vocab.doc import Example

    ref = Doc(vocab, ["children"], tags=["NNS"], lemmas=["child"])
    eg = Example.from_doc(ref)
    eg.predicted.tags = ["NNS"]
    lem([eg.predicted])
    scores = score_examples([eg], ["lemmatizer"])
    assert scores["lemma_acc"] == 1.0


def test_en_core_web_sm_shaped_pipeline():
    """The REAL en_core_web_sm component lineup — tok2vec, tagger, parser,
    ner, attribute_ruler, lemmatizer — builds from a config, trains the
    statistical components, runs the rule components in predict order
    (attribute_ruler AFTER the tagger, lemmatizer last), and round-trips
    through to_disk/load."""
    import tempfile

    import torch

    import spacy_ray_amd
    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.pipeline.language import init_nlp

    cfg_text = """
[nlp]
lang = "en"
pipeline = ["tok2vec", "tagger", "parser", "ner", "attribute_ruler", "lemmatizer"]

[components]

[components.tok2vec]
factory = "tok2vec"

[components.tok2vec.model]
@architectures = "spacy.HashEmbedCNN.v2"
width = 64
depth = 2
embed_size = 500
window_size = 1
maxout_pieces = 3
subword_features = true
pretrained_vectors = null

[components.tagger]
factory = "tagger"

[components.tagger.model]
@architectures = "spacy.Tagger.v2"

[components.tagger.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64

[components.parser]
factory = "parser"

[components.parser.model]
@architectures = "spacy.TransitionBasedParser.v2"
state_type = "parser"

[components.parser.model.tok2vec]
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

[components.attribute_ruler]
factory = "attribute_ruler"

[components.lemmatizer]
factory = "lemmatizer"
mode = "rule"

[corpora]

[corpora.train]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 64
words_per_doc = 12
vocab_size = 150
n_tags = 10
seed = 0

[corpora.dev]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 16
seed = 1

[training]
seed = 0
train_corpus = "corpora.train"
dev_corpus = "corpora.dev"

[training.optimizer]
@optimizers = "Adam.v1"
learn_rate = 0.001
"""
    torch.manual_seed(0)
    nlp = init_nlp(Config.from_str(cfg_text), device="cpu", sample_size=16)
    nlp.get_pipe("attribute_ruler").add_patterns(
        [{"patterns": [[{"ORTH": "w1"}]], "attrs": {"TAG": "SPECIAL"}}])
    docs = make_synthetic_docs(nlp.vocab, n_docs=6, words_per_doc=10,
                               vocab_size=40, n_tags=10, n_deps=5,
                               n_ent_types=2, seed=2)
    outs = nlp.predict_docs([d.copy_unannotated() for d in docs])
    for d in outs:
        assert d.tags and d.heads is not None and d.ents is not None
        assert d.lemmas is not None and all(d.lemmas)
        for w, t in zip(d.words, d.tags):
            if w == "w1":
                assert t == "SPECIAL"  # ruler ran after the tagger
    with tempfile.TemporaryDirectory() as td:
        nlp.to_disk(td)
        nlp2 = spacy_ray_amd.load(td, device="cpu")
        outs2 = nlp2.predict_docs([d.copy_unannotated() for d in docs])
        assert [d.lemmas for d in outs2] == [d.lemmas for d in outs]
        assert [d.tags for d in outs2] == [d.tags for d in outs]


def test_entity_ruler_after_ner_annotation_order():
    """Annotations land in pipeline order: an entity_ruler placed AFTER
    ner fills O tokens without being clobbered by ner's decode writes
    (the spaCy ruler-after-ner pattern)."""
    import torch

    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.pipeline.language import init_nlp

    cfg = Config.from_disk("examples/configs/en_core_cnn.cfg")
    torch.manual_seed(0)
    nlp = init_nlp(cfg, device="cpu", sample_size=16)
    # append a ruler AFTER ner
    from spacy_ray_amd.pipeline.ruler import EntityRulerPipe

    ruler = EntityRulerPipe("entity_ruler")
    nlp.add_pipe("entity_ruler", ruler)
    docs = make_synthetic_docs(nlp.vocab, n_docs=4, words_per_doc=10,
                               vocab_size=50, n_tags=10, n_deps=5,
                               n_ent_types=2, seed=6)
    # pick a word the (untrained) ner labels O in at least one doc
    base = nlp.predict_docs([d.copy_unannotated() for d in docs])
    target = None
    for d in base:
        for w, e in zip(d.words, d.ents):
            if e == "O":
                target = w
                break
        if target:
            break
    assert target is not None
    ruler.add_patterns([{"label": "RULED", "pattern": target}])
    outs = nlp.predict_docs([d.copy_unannotated() for d in docs])
    ruled = [e for d in outs for w, e in zip(d.words, d.ents)
             if w == target]
    assert any(e == "U-RULED" for e in ruled), ruled


def test_spancat_checkpoint_roundtrip(tmp_path):
    """spancat labels/spans_key/threshold survive to_disk/load and
    predictions are identical."""
    import numpy as np
    import torch

    import spacy_ray_amd
    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.pipeline.language import init_nlp

    cfg_text = TEXTCAT_CFG.replace(
        'pipeline = ["tok2vec", "textcat", "senter"]',
        'pipeline = ["tok2vec", "spancat"]',
    ).replace("""[components.textcat]
factory = "textcat"

[components.textcat.model]
@architectures = "spacy.TextCatReduce.v1"
exclusive_classes = true

[components.textcat.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64

[components.senter]
factory = "senter"

[components.senter.model]
@architectures = "spacy.Tagger.v2"

[components.senter.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64""",
"""[components.spancat]
factory = "spancat"
spans_key = "myspans"
max_ngram = 2
threshold = 0.4

[components.spancat.model]
@architectures = "spacy.SpanCategorizer.v1"

[components.spancat.model.tok2vec]
@architectures = "spacy.Tok2VecListener.v1"
width = 64""")
    torch.manual_seed(0)
    nlp = init_nlp(Config.from_str(cfg_text), device="cpu", sample_size=8)
    pipe = nlp.get_pipe("spancat")
    pipe.labels = ["A", "B"]
    pipe.label2id = {"A": 0, "B": 1}
    pipe.module = None
    pipe.initialize([], "cpu")
    docs = make_synthetic_docs(nlp.vocab, n_docs=4, words_per_doc=8,
                               vocab_size=40, n_tags=5, n_deps=5,
                               n_ent_types=2, seed=8)
    outs1 = nlp.predict_docs([d.copy_unannotated() for d in docs])
    nlp.to_disk(str(tmp_path))
    nlp2 = spacy_ray_amd.load(str(tmp_path), device="cpu")
    p2 = nlp2.get_pipe("spancat")
    assert p2.labels == ["A", "B"] and p2.spans_key == "myspans"
    assert p2.max_ngram == 2 and abs(p2.threshold - 0.4) < 1e-9
    outs2 = nlp2.predict_docs([d.copy_unannotated() for d in docs])
    assert [d.spans for d in outs2] == [d.spans for d in outs1]


def test_annotating_components_with_transition_and_rule_pipes():
    """training.annotating_components: listed pipes set predictions on
    eg.predicted before losses are computed — covers the transition-pipe
    branch (greedy + beam) and rule pipes."""
    import torch

    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.pipeline.language import init_nlp
    from spacy_ray_amd.vocab.doc import Example

    cfg = Config.from_disk("examples/configs/en_core_cnn.cfg")
    torch.manual_seed(0)
    nlp = init_nlp(cfg, device="cpu", sample_size=16)
    nlp._annotating = ["parser", "ner"]
    docs = make_synthetic_docs(nlp.vocab, n_docs=6, words_per_doc=8,
                               vocab_size=50, n_tags=10, n_deps=5,
                               n_ent_types=2, seed=9)
    examples = [Example.from_doc(d) for d in docs]
    total, losses = nlp.forward_loss(examples, drop=0.0)
    assert all(eg.predicted.heads is not None for eg in examples)
    assert all(eg.predicted.ents is not None for eg in examples)
    # beam branch of the annotating path
    nlp.get_pipe("parser").beam_width = 2
    examples2 = [Example.from_doc(d) for d in docs]
    nlp.forward_loss(examples2, drop=0.0)
    assert all(eg.predicted.heads is not None for eg in examples2)
    nlp.get_pipe("parser").beam_width = 1
