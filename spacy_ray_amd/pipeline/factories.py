"""Pipe factories (the ``factory = "..."`` names in [components.*] blocks)."""
from __future__ import annotations

from spacy_ray_amd.config.registry import registry
from .pipes import NerPipe, ParserPipe, TaggerPipe, Tok2VecPipe


@registry.factories("tok2vec")
def make_tok2vec_pipe(name: str, model):
    return Tok2VecPipe(name, model)


@registry.factories("transformer")
def make_transformer_pipe(name: str, model):
    # the transformer IS the shared tok2vec of the pipeline (listener heads
    # consume its [T, width] output exactly like the CNN path)
    return Tok2VecPipe(name, model)


def _with_labels(pipe, labels):
    """Explicit label list from the component config (spaCy's `labels` /
    initialize-labels contract): skips corpus discovery entirely."""
    if labels:
        pipe.labels = list(labels)
        pipe.label2id = {t: i for i, t in enumerate(pipe.labels)}
    return pipe


@registry.factories("tagger")
def make_tagger_pipe(name: str, model, labels=None):
    return _with_labels(TaggerPipe(name, model), labels)


@registry.factories("parser")
def make_parser_pipe(name: str, model, labels=None, use_break: bool = False,
                     beam_width: int = 1):
    return _with_labels(
        ParserPipe(name, model, use_break=use_break, beam_width=beam_width),
        labels)


@registry.factories("ner")
def make_ner_pipe(name: str, model, labels=None, beam_width: int = 1):
    return _with_labels(NerPipe(name, model, beam_width=beam_width), labels)


@registry.factories("textcat")
def make_textcat_pipe(name: str, model, labels=None):
    from .pipes import TextcatPipe

    return _with_labels(TextcatPipe(name, model, exclusive_classes=True), labels)


@registry.factories("textcat_multilabel")
def make_textcat_multilabel_pipe(name: str, model, labels=None):
    from .pipes import TextcatPipe

    return _with_labels(TextcatPipe(name, model, exclusive_classes=False), labels)


@registry.factories("senter")
def make_senter_pipe(name: str, model):
    from .pipes import SenterPipe

    return SenterPipe(name, model)


@registry.factories("entity_ruler")
def make_entity_ruler_pipe(name: str, model=None, overwrite_ents: bool = False,
                           patterns=None):
    from .ruler import EntityRulerPipe

    return EntityRulerPipe(name, model, overwrite_ents=overwrite_ents,
                           patterns=patterns)


@registry.factories("morphologizer")
def make_morphologizer_pipe(name: str, model, labels=None):
    from .pipes import MorphologizerPipe

    return _with_labels(MorphologizerPipe(name, model), labels)


@registry.factories("spancat")
def make_spancat_pipe(name: str, model, spans_key: str = "sc",
                      max_ngram: int = 3, threshold: float = 0.5,
                      labels=None):
    from .pipes import SpancatPipe

    return _with_labels(
        SpancatPipe(name, model, spans_key=spans_key, max_ngram=max_ngram,
                    threshold=threshold), labels)


@registry.factories("attribute_ruler")
def make_attribute_ruler_pipe(name: str, model=None, patterns=None):
    from .attr_ruler import AttributeRulerPipe

    return AttributeRulerPipe(name, model, patterns=patterns)


@registry.factories("lemmatizer")
def make_lemmatizer_pipe(name: str, model=None, mode: str = "rule",
                         lookups=None, overwrite: bool = False):
    from .lemmatizer import LemmatizerPipe

    return LemmatizerPipe(name, model, mode=mode, lookups=lookups,
                          overwrite=overwrite)
