"""Config parsing / interpolation / overrides / registry resolution."""
import pytest

from spacy_ray_amd.config.config import Config, parse_config_overrides, resolve

CFG = """
[paths]
train = null
dev = "dev.spacy"

[nlp]
lang = "en"
pipeline = ["tok2vec", "tagger"]

[section]
number = 3
flt = 0.5
flag = true
items = [1, 2, 3]
ref = ${section.number}
embedded = "w=${section.number}"
coloned = ${nlp:lang}

[section.sub]
x = 1

[training.optimizer]
@optimizers = "Adam.v1"
learn_rate = 0.01
"""


def test_parse_types():
    cfg = Config.from_str(CFG)
    s = cfg["section"]
    assert s["number"] == 3
    assert s["flt"] == 0.5
    assert s["flag"] is True
    assert s["items"] == [1, 2, 3]
    assert cfg["paths"]["train"] is None
    assert cfg["nlp"]["pipeline"] == ["tok2vec", "tagger"]
    assert cfg["section"]["sub"]["x"] == 1


def test_interpolation():
    cfg = Config.from_str(CFG).interpolate()
    assert cfg["section"]["ref"] == 3  # whole-value ref keeps type
    assert cfg["section"]["embedded"] == "w=3"
    assert cfg["section"]["coloned"] == "en"


def test_roundtrip():
    cfg = Config.from_str(CFG)
    cfg2 = Config.from_str(cfg.to_str())
    assert cfg == cfg2


def test_overrides():
    overrides = parse_config_overrides(["--section.number", "7", "--nlp.lang=de"])
    cfg = Config.from_str(CFG, overrides=overrides)
    assert cfg["section"]["number"] == 7
    assert cfg["nlp"]["lang"] == "de"


def test_resolve_registry_block():
    cfg = Config.from_str(CFG).interpolate()
    out = resolve(cfg["training"])
    from spacy_ray_amd.train.optimizer import AdamSpec

    assert isinstance(out["optimizer"], AdamSpec)
    assert out["optimizer"].lr(0) == 0.01


def test_bad_override():
    with pytest.raises(ValueError):
        parse_config_overrides(["positional"])


def test_interpolation_cycle_detected():
    cfg = Config.from_str("""
[a]
x = ${b.y}

[b]
y = ${a.x}
""")
    with pytest.raises(ValueError):
        cfg.interpolate()


def test_registry_unknown_name_errors():
    from spacy_ray_amd.config.registry import registry

    registry.ensure_populated()
    with pytest.raises(KeyError):
        registry.architectures.get("no.such.arch.v1")
    with pytest.raises(KeyError):
        registry.get_registry("nosuchregistry")
    with pytest.raises(ValueError):
        resolve({"@optimizers": "Adam.v1", "@schedules": "constant.v1"})


def test_training_schema_validation():
    from spacy_ray_amd.config.schemas import ConfigSchemaTraining

    out = resolve(
        {"dropout": 0.2, "max_steps": 5, "custom_extra": 1,
         "optimizer": {"@optimizers": "Adam.v1", "learn_rate": 0.01}},
        schema=ConfigSchemaTraining,
    )
    assert out["dropout"] == 0.2
    assert out["accumulate_gradient"] == 1  # default filled
    assert out["custom_extra"] == 1         # extras pass through
    from spacy_ray_amd.train.optimizer import AdamSpec

    assert isinstance(out["optimizer"], AdamSpec)
    with pytest.raises(Exception):
        resolve({"accumulate_gradient": 0}, schema=ConfigSchemaTraining)
    with pytest.raises(Exception):
        resolve({"dropout": 1.5}, schema=ConfigSchemaTraining)
