"""Registered model architectures (the @architectures names spaCy configs
reference — SURVEY.md §5.6; resolution happens when Language builds pipes).

Architecture functions return *model specs* (factory closures) because output
dims (labels) are only known at pipe.initialize() time, mirroring spaCy's
shape-inference-on-sample initialization (`/root/reference/spacy_ray/
worker.py:91` init_nlp contract)."""
from __future__ import annotations

from typing import List, Optional

from spacy_ray_amd.config.registry import registry
from .tok2vec import MaxoutWindowEncoder, MultiHashEmbed, Tok2Vec
from .parser_model import TransitionModel


class ModelSpec:
    """A deferred model constructor: call .build() for the nn.Module."""

    def __init__(self, build, width: Optional[int] = None, kind: str = ""):
        self._build = build
        self.width = width
        self.kind = kind

    def build(self):
        return self._build()


@registry.architectures("spacy.MultiHashEmbed.v2")
def make_multi_hash_embed(
    width: int,
    attrs: List[str] = ("NORM", "PREFIX", "SUFFIX", "SHAPE"),
    rows: List[int] = (5000, 2500, 2500, 2500),
    include_static_vectors: bool = False,
    seed: int = 0,
):
    if include_static_vectors:
        raise NotImplementedError("static vectors are not supported (no pretrained vectors offline)")
    return ModelSpec(lambda: MultiHashEmbed(width, rows=list(rows), attrs=list(attrs), seed=seed),
                     width=width, kind="embed")


@registry.architectures("spacy.MaxoutWindowEncoder.v2")
def make_maxout_window_encoder(
    width: int, depth: int = 4, window_size: int = 1, maxout_pieces: int = 3
):
    return ModelSpec(
        lambda: MaxoutWindowEncoder(width, depth=depth, window_size=window_size,
                                    maxout_pieces=maxout_pieces),
        width=width, kind="encode",
    )


@registry.architectures("spacy.Tok2Vec.v2")
def make_tok2vec(embed: ModelSpec, encode: ModelSpec):
    return ModelSpec(lambda: Tok2Vec(embed.build(), encode.build()),
                     width=encode.width, kind="tok2vec")


@registry.architectures("spacy.HashEmbedCNN.v2")
def make_hash_embed_cnn(
    width: int = 96,
    depth: int = 4,
    embed_size: int = 2000,
    window_size: int = 1,
    maxout_pieces: int = 3,
    subword_features: bool = True,
    pretrained_vectors: Optional[str] = None,
):
    rows = [embed_size, embed_size // 2, embed_size // 2, embed_size // 2] if subword_features else [embed_size]
    attrs = ["NORM", "PREFIX", "SUFFIX", "SHAPE"] if subword_features else ["NORM"]

    def build():
        return Tok2Vec(
            MultiHashEmbed(width, rows=rows, attrs=attrs),
            MaxoutWindowEncoder(width, depth=depth, window_size=window_size,
                                maxout_pieces=maxout_pieces),
        )

    return ModelSpec(build, width=width, kind="tok2vec")


@registry.architectures("spacy-transformers.TransformerModel.v3")
def make_transformer_model(
    name: str = "roberta-base",
    get_spans=None,
    tokenizer_config: Optional[dict] = None,
    transformer_config: Optional[dict] = None,
    window: int = 128,
    stride: int = 96,
    attn_implementation: str = "sdpa",
    subwords: str = "bpe",
    bpe_vocab_size: int = 8000,
):
    tk = tokenizer_config or {}

    def build():
        from .transformer import TransformerTok2Vec

        return TransformerTok2Vec(name=name, window=window, stride=stride,
                                  transformer_config=transformer_config,
                                  attn_implementation=attn_implementation,
                                  subwords=tk.get("subwords", subwords),
                                  bpe_vocab_size=tk.get("bpe_vocab_size", bpe_vocab_size),
                                  tokenizer_path=tk.get("tokenizer_path"))

    tc = transformer_config or {}
    width = tc.get("hidden_size", 768 if name == "roberta-base" else 64)
    return ModelSpec(build, width=width, kind="tok2vec")


@registry.architectures("spacy.Tok2VecListener.v1")
def make_tok2vec_listener(width: int, upstream: str = "*"):
    return ModelSpec(lambda: None, width=width, kind="listener")


@registry.architectures("spacy.Tagger.v2")
def make_tagger_model(tok2vec: ModelSpec, nO: Optional[int] = None, normalize: bool = False):
    spec = ModelSpec(lambda: None, width=tok2vec.width, kind="tagger")
    # a full tok2vec block (not a listener) => the pipe owns its encoder
    # (spaCy's embedded-tok2vec configuration, e.g. standalone taggers)
    spec.embedded_tok2vec = tok2vec if tok2vec.kind == "tok2vec" else None
    return spec


@registry.architectures("spacy.TransitionBasedParser.v2")
def make_transition_parser_model(
    tok2vec: ModelSpec,
    state_type: str = "parser",
    extra_state_tokens: bool = False,
    hidden_width: int = 64,
    maxout_pieces: int = 2,
    use_upper: bool = True,
    nO: Optional[int] = None,
):
    nF = 13 if state_type == "parser" else 6

    def build():
        return TransitionModel(tok2vec.width, hidden_width=hidden_width, nF=nF)

    spec = ModelSpec(build, width=tok2vec.width, kind=f"transition:{state_type}")
    spec.nF = nF
    spec.hidden_width = hidden_width
    spec.embedded_tok2vec = tok2vec if tok2vec.kind == "tok2vec" else None
    return spec


@registry.architectures("spacy.TextCatCNN.v2")
def make_textcat_cnn_model(tok2vec: ModelSpec, exclusive_classes: bool = True,
                           nO: Optional[int] = None):
    spec = ModelSpec(lambda: None, width=tok2vec.width, kind="textcat")
    spec.embedded_tok2vec = tok2vec if tok2vec.kind == "tok2vec" else None
    return spec


@registry.architectures("spacy.TextCatReduce.v1")
def make_textcat_reduce_model(tok2vec: ModelSpec,
                              exclusive_classes: bool = True,
                              use_reduce_mean: bool = True, **_ignored):
    spec = ModelSpec(lambda: None, width=tok2vec.width, kind="textcat")
    spec.embedded_tok2vec = tok2vec if tok2vec.kind == "tok2vec" else None
    return spec


@registry.architectures("spacy.SpanCategorizer.v1")
def make_spancat_model(tok2vec: ModelSpec, **_ignored):
    spec = ModelSpec(lambda: None, width=tok2vec.width, kind="spancat")
    spec.embedded_tok2vec = tok2vec if tok2vec.kind == "tok2vec" else None
    return spec
