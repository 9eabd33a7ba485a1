"""Thinc-msgpack component bytes (spaCy checkpoint interop layer)."""
import msgpack
import numpy as np

from spacy_ray_amd.data.thinc_serde import (
    _decode_array, _encode_array, model_to_thinc_bytes, thinc_bytes_to_model)


def test_msgpack_numpy_encoding_roundtrip():
    a = np.arange(12, dtype=np.float32).reshape(3, 4)
    enc = _encode_array(a)
    assert enc[b"nd"] is True and enc[b"type"] == "<f4"
    b = _decode_array(enc)
    assert np.array_equal(a, b)


def test_model_bytes_layout_and_roundtrip():
    nodes = [
        ("maxout", {"W": np.random.rand(8, 4).astype("f"), "b": np.zeros(8, "f")}),
        ("layernorm", {"G": np.ones(8, "f"), "b": np.zeros(8, "f")}),
    ]
    data = model_to_thinc_bytes(nodes)
    msg = msgpack.unpackb(data, raw=False, strict_map_key=False)
    assert set(msg) == {"nodes", "attrs", "params", "shims"}
    assert msg["nodes"][0]["name"] == "maxout"
    assert msg["nodes"][0]["index"] == 0
    back = thinc_bytes_to_model(data)
    assert back[0][0] == "maxout"
    assert np.allclose(back[0][1]["W"], nodes[0][1]["W"])


def test_component_thinc_bytes_written_and_reloadable(tmp_path):
    """to_disk writes <component>/model (Thinc layout); a fresh pipeline can
    restore params from ONLY those bytes (safetensors removed)."""
    import torch

    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.pipeline.language import init_nlp
    from tests.test_pipeline import TAGGER_CFG

    cfg = Config.from_str(TAGGER_CFG)
    torch.manual_seed(0)
    nlp = init_nlp(cfg)
    out = tmp_path / "model"
    nlp.to_disk(out)
    assert (out / "tagger" / "model").exists()
    assert (out / "tok2vec" / "model").exists()
    assert (out / "tokenizer").exists()
    # drop safetensors -> force the thinc-bytes load path
    for p in out.rglob("model.safetensors"):
        p.unlink()
    torch.manual_seed(1)  # different init; loading must overwrite it
    nlp2 = init_nlp(cfg)
    nlp2.from_disk(out)
    t1 = nlp.get_pipe("tok2vec").module
    t2 = nlp2.get_pipe("tok2vec").module
    for (n1, p1), (n2, p2) in zip(t1.named_parameters(), t2.named_parameters()):
        assert n1 == n2
        assert torch.allclose(p1, p2, atol=1e-6), n1
    h1 = nlp.get_pipe("tagger").module
    h2 = nlp2.get_pipe("tagger").module
    assert torch.allclose(h1.output.weight, h2.output.weight)


def test_thinc_bytes_roundtrip_new_pipe_types():
    """Thinc-msgpack component bytes restore params for every new
    trainable pipe type (textcat/senter/morphologizer/spancat)."""
    import torch

    from spacy_ray_amd.data.thinc_serde import (
        component_to_thinc_nodes, load_thinc_nodes_into_component,
        model_to_thinc_bytes)
    from spacy_ray_amd.pipeline.pipes import (MorphologizerPipe, SenterPipe,
                                              SpancatPipe, TextcatPipe)

    class Spec:
        width = 32
        embedded_tok2vec = None

    for cls in (TextcatPipe, SenterPipe, MorphologizerPipe, SpancatPipe):
        p = cls("x", Spec())
        p.labels = ["A", "B"]
        p.label2id = {"A": 0, "B": 1}
        p.initialize([], "cpu")
        with torch.no_grad():
            for t in p.module.parameters():
                t.add_(torch.randn_like(t))
        blob = model_to_thinc_bytes(component_to_thinc_nodes(p))
        ref = {k: v.clone() for k, v in p.module.state_dict().items()}
        with torch.no_grad():
            for t in p.module.parameters():
                t.zero_()
        assert load_thinc_nodes_into_component(p, blob) > 0
        for k, v in p.module.state_dict().items():
            assert torch.allclose(v, ref[k]), (cls.__name__, k)
