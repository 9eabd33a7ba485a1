"""spacy_ray_amd: MI355X-native distributed training engine with the
capability surface of explosion/spacy-ray.

Layer map (idiomatic MI355X rebuild of SURVEY.md §1):
  cli       — `spacy-mi ray train` entry (flag surface of train_cli.py:26-37)
  config    — confection-compatible .cfg parsing + registry resolution
  vocab     — StringStore, Doc, attr extraction (murmur-hashed)
  data      — corpora (synthetic + DocBin), word-count batcher
  models    — tok2vec CNN / transformer, tagger, transition parser (torch)
  ops       — hand-written gfx950 HIP kernels + pure-torch references
  pipeline  — Language container + trainable pipes, to_disk/from_disk
  parallel  — 1-proc-per-GPU comm engine: bucketed reduce-scatter +
              sharded Adam + all-gather on RCCL over xGMI (ZeRO-1)
  train     — train_while_improving-contract loop, loggers, scoring
"""
__version__ = "0.1.0"


def load(model_path, device: str = "cpu"):
    """Load a trained pipeline directory (the `spacy.load` convenience):
    ``nlp = spacy_ray_amd.load("/path/model-best", device="cuda:0")``."""
    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.pipeline.language import build_nlp

    config = Config.from_disk(f"{model_path}/config.cfg")
    nlp = build_nlp(config, device=device)
    nlp.from_disk(model_path)
    return nlp
