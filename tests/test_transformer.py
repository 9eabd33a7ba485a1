"""Transformer tok2vec path (config #4 shape) on CPU with a tiny encoder."""
import pytest
import torch

from spacy_ray_amd.config.config import Config, resolve, resolve_dot_names
from spacy_ray_amd.pipeline.language import init_nlp
from spacy_ray_amd.train.stepper import SimpleStepper

TRF_CFG = """
[nlp]
lang = "en"
pipeline = ["transformer", "tagger"]

[components]

[components.transformer]
factory = "transformer"

[components.transformer.model]
@architectures = "spacy-transformers.TransformerModel.v3"
name = "tiny"
window = 16
stride = 12

[components.transformer.model.transformer_config]
vocab_size = 512
hidden_size = 32
num_hidden_layers = 1
num_attention_heads = 2
intermediate_size = 64

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
n_docs = 50
words_per_doc = 30
vocab_size = 100
n_tags = 5
seed = 0

[corpora.dev]
@readers = "spacy-mi.SyntheticCorpus.v1"
n_docs = 20
words_per_doc = 30
vocab_size = 100
n_tags = 5
seed = 1
shuffle = false

[training]
seed = 0
max_steps = 5
train_corpus = "corpora.train"
dev_corpus = "corpora.dev"

[training.batcher]
@batchers = "spacy.batch_by_words.v1"
size = 300

[training.optimizer]
@optimizers = "Adam.v1"
learn_rate = 0.001

[training.score_weights]
tag_acc = 1.0
"""


def test_transformer_pipeline_trains_cpu():
    cfg = Config.from_str(TRF_CFG)
    nlp = init_nlp(cfg, sample_size=16)
    icfg = cfg.interpolate()
    T = resolve(icfg["training"], validate=False)
    (train_corpus,) = resolve_dot_names(icfg, [T["train_corpus"]])
    examples = []
    for eg in train_corpus(nlp):
        examples.append(eg)
        if len(examples) >= 8:
            break
    stepper = SimpleStepper(nlp, T["optimizer"])
    losses = {}
    for _ in range(3):
        stepper.accumulate(examples, drop=0.0, losses=losses)
        stepper.apply_step()
    assert losses["tagger"] > 0
    # windowing: docs of 30 words with window 16 produce overlapping spans;
    # output must still be one vector per token
    from spacy_ray_amd.models.batch import TokenBatch

    batch = TokenBatch([eg.predicted for eg in examples], nlp.device)
    Y = nlp.tok2vec.forward(batch)
    assert Y.shape == (batch.n_tokens, 32)
    assert torch.isfinite(Y).all()


def test_mixed_treebank_corpus():
    from spacy_ray_amd.config.registry import registry

    registry.ensure_populated()
    corpus = registry.readers.get("spacy-mi.MixedTreebankCorpus.v1")(
        n_treebanks=3, docs_per_treebank=5, words_per_doc=10,
        vocab_size_per_treebank=50, n_tags=17, n_deps=37, seed=0)

    class FakeNlp:
        from spacy_ray_amd.vocab.doc import Vocab

        vocab = Vocab("xx")

    egs = list(corpus(FakeNlp))
    assert len(egs) == 15
    prefixes = {eg.reference.words[0].split(":")[0] for eg in egs}
    assert len(prefixes) == 3  # mixed languages present


def test_transformer_long_doc_window_stitching():
    """Docs longer than the window split into strided spans; overlapping
    positions average, output rows == token count, and a long doc's
    embedding for shared tokens stays finite and scaled correctly."""
    import torch

    from spacy_ray_amd.models.transformer import TransformerTok2Vec
    from spacy_ray_amd.models.batch import TokenBatch
    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.vocab.doc import Vocab

    vocab = Vocab()
    t2v = TransformerTok2Vec(
        name="tiny-test", window=8, stride=5,
        transformer_config={"hidden_size": 16, "num_hidden_layers": 1,
                            "num_attention_heads": 2,
                            "intermediate_size": 32,
                            "max_position_embeddings": 64,
                            "vocab_size": 1000},
    )
    docs = make_synthetic_docs(vocab, n_docs=3, words_per_doc=20,
                               vocab_size=50, n_tags=5, n_deps=3,
                               n_ent_types=1, seed=3)
    total = sum(len(d) for d in docs)
    batch = TokenBatch(docs, torch.device("cpu"))
    out = t2v(batch)
    assert out.shape[0] >= total  # padded rows allowed at the tail
    assert torch.isfinite(out[:total]).all()
    # windows of 8 with stride 5: a 20-token doc needs spans covering all
    # 20 positions
    spans = t2v._windows([20])
    covered = set()
    for a, b in spans:
        covered.update(range(a, b))
    assert covered == set(range(20))
    # stride < window => interior tokens appear in overlapping windows
    assert len(spans) == 4  # starts 0,5,10,15 with window 8 over 20


def test_bpe_subwords_alignment_and_checkpoint(tmp_path):
    """Default subwords='bpe': real byte-level BPE trained at init; output
    stays [n_words, width] via word-alignment pooling; the trained
    tokenizer round-trips through the checkpoint."""
    from spacy_ray_amd.vocab.doc import Doc, Example

    cfg = Config.from_str(TRF_CFG)
    nlp = init_nlp(cfg)
    trf = nlp.get_pipe("transformer").module
    assert trf.bpe is not None and trf.bpe.tok is not None
    # a word the trainer never saw still tokenizes (byte-level has no OOV)
    doc = Doc(nlp.vocab, ["w1", "zzzzunseenzzzz", "w3"])
    ids, wid = trf.bpe.encode_doc(doc)
    assert len(ids) >= 3 and wid.max() == 2
    from spacy_ray_amd.models.batch import TokenBatch

    tb = TokenBatch([doc], torch.device("cpu"))
    out = trf(tb)
    assert out.shape == (tb.n_tokens, trf.width)
    # checkpoint: the serialized tokenizer restores identical segmentation
    out_dir = tmp_path / "m"
    nlp.to_disk(out_dir)
    nlp2 = init_nlp(cfg)
    nlp2.from_disk(out_dir)
    trf2 = nlp2.get_pipe("transformer").module
    d2 = Doc(nlp2.vocab, ["w1", "zzzzunseenzzzz", "w3"])
    ids2, wid2 = trf2.bpe.encode_doc(d2)
    assert ids2.tolist() == ids.tolist()
    assert wid2.tolist() == wid.tolist()


def test_hash_subwords_fallback():
    cfg_text = TRF_CFG.replace(
        'stride = 12', 'stride = 12\nsubwords = "hash"')
    nlp = init_nlp(Config.from_str(cfg_text))
    trf = nlp.get_pipe("transformer").module
    assert trf.bpe is None
    from spacy_ray_amd.models.batch import TokenBatch
    from spacy_ray_amd.vocab.doc import Doc

    doc = Doc(nlp.vocab, ["a", "b", "c", "d"])
    out = trf(TokenBatch([doc], torch.device("cpu")))
    assert out.shape[0] == 4


def test_window_length_bucketing_equivalent_outputs():
    """Length-bucketed window batching (each window pads to its bucket's L
    instead of the batch-global max) must produce the same outputs as a
    single global pad length — padding is fully masked."""
    import re

    import numpy as np

    from spacy_ray_amd.data.corpus import make_synthetic_docs
    from spacy_ray_amd.models.batch import TokenBatch
    from spacy_ray_amd.vocab.doc import Doc

    cfg_text = TRF_CFG.replace("window = 16", "window = 128").replace(
        "stride = 12", "stride = 96")
    torch.manual_seed(0)
    nlp = init_nlp(Config.from_str(cfg_text))
    trf = nlp.get_pipe("transformer").module
    trf.eval()
    docs = make_synthetic_docs(nlp.vocab, n_docs=20, words_per_doc=12,
                               vocab_size=200, n_tags=5, n_deps=5,
                               n_ent_types=2, seed=2)
    docs.append(Doc(nlp.vocab, [f"w{i % 50}" for i in range(70)]))
    tb = TokenBatch(docs, torch.device("cpu"))
    with torch.no_grad():
        y_bucketed = trf(tb)
        trf.BUCKETS = ()  # single global bucket = pre-bucketing behavior
        y_single = trf(tb)
    assert torch.allclose(y_bucketed, y_single, atol=2e-5)
