"""Corpus converters: CoNLL-U / IOB -> DocBin."""
import subprocess
import sys
from pathlib import Path

from spacy_ray_amd.data.convert import iob_to_biluo, read_conllu, read_iob

REPO = Path(__file__).resolve().parent.parent

CONLLU = """\
# sent_id = 1
# text = The dog barks
1\tThe\tthe\tDET\tDT\t_\t2\tdet\t_\t_
2\tdog\tdog\tNOUN\tNN\t_\t3\tnsubj\t_\t_
3\tbarks\tbark\tVERB\tVBZ\t_\t0\troot\t_\t_

1-2\tcannot\t_\t_\t_\t_\t_\t_\t_\t_
1\tcan\tcan\tAUX\tMD\t_\t0\troot\t_\t_
2\tnot\tnot\tPART\tRB\t_\t1\tadvmod\t_\t_
2.1\tghost\t_\t_\t_\t_\t_\t_\t_\t_
"""

IOB = """\
-DOCSTART- -X- O O

EU NNP I-ORG
rejects VBZ O
German JJ I-MISC
call NN O

Peter NNP B-PER
Blackburn NNP I-PER
"""


def test_read_conllu():
    docs = read_conllu(CONLLU)
    assert len(docs) == 2
    d = docs[0]
    assert d.words == ["The", "dog", "barks"]
    assert d.tags == ["DET", "NOUN", "VERB"]
    assert list(d.heads) == [1, 2, -1]
    assert d.deps == ["det", "nsubj", "root"]
    d2 = docs[1]  # multiword range and empty node skipped
    assert d2.words == ["can", "not"]
    assert list(d2.heads) == [-1, 0]


def test_read_conllu_xpos():
    docs = read_conllu(CONLLU, tag_col="xpos")
    assert docs[0].tags == ["DT", "NN", "VBZ"]


def test_iob_to_biluo():
    # IOB1: I- opens a span
    assert iob_to_biluo(["I-ORG", "O", "I-MISC", "O"]) == ["U-ORG", "O", "U-MISC", "O"]
    # IOB2: B- opens, I- continues
    assert iob_to_biluo(["B-PER", "I-PER"]) == ["B-PER", "L-PER"]
    assert iob_to_biluo(["B-PER", "I-PER", "I-PER", "O"]) == ["B-PER", "I-PER", "L-PER", "O"]
    # adjacent spans, label change
    assert iob_to_biluo(["B-A", "B-A"]) == ["U-A", "U-A"]
    assert iob_to_biluo(["I-A", "I-B"]) == ["U-A", "U-B"]
    assert iob_to_biluo([]) == []


def test_read_iob():
    docs = read_iob(IOB)
    assert len(docs) == 2
    assert docs[0].words == ["EU", "rejects", "German", "call"]
    assert docs[0].ents == ["U-ORG", "O", "U-MISC", "O"]
    assert docs[0].tags == ["NNP", "VBZ", "JJ", "NN"]
    assert docs[1].ents == ["B-PER", "L-PER"]


def test_convert_cli_roundtrip(tmp_path):
    src = tmp_path / "sample.conllu"
    src.write_text(CONLLU)
    out = tmp_path / "sample.spacy"
    r = subprocess.run(
        [sys.executable, "-m", "spacy_ray_amd.cli.main", "convert",
         str(src), str(out)],
        cwd=str(REPO), capture_output=True, text=True, timeout=120,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    from spacy_ray_amd.data.docbin import DocBin
    from spacy_ray_amd.vocab.doc import Vocab

    docs = list(DocBin.from_disk(out, Vocab()).get_docs(Vocab()))
    assert len(docs) == 2 and docs[0].words == ["The", "dog", "barks"]


def test_convert_then_train_end_to_end(tmp_path):
    """The full user data path: CoNLL-U file -> spacy-mi convert -> DocBin ->
    spacy.Corpus.v1 reader -> training runs and checkpoints."""
    import random

    rng = random.Random(0)
    lex = [f"w{i}" for i in range(30)]
    sents = []
    for _ in range(60):
        n = rng.randint(3, 8)
        rows = []
        for i in range(1, n + 1):
            w = lex[rng.randrange(len(lex))]
            tag = "NOUN" if w < "w5" else "VERB"
            head = 0 if i == 1 else i - 1  # chain tree rooted at token 1
            dep = "root" if head == 0 else "dep"
            rows.append(f"{i}\t{w}\t_\t{tag}\t_\t_\t{head}\t{dep}\t_\t_")
        sents.append("\n".join(rows))
    src = tmp_path / "train.conllu"
    src.write_text("\n\n".join(sents) + "\n")

    from spacy_ray_amd.data.convert import convert_file

    train_bin = tmp_path / "train.spacy"
    n = convert_file(src, train_bin)
    assert n == 60

    from tests.test_pipeline import TAGGER_CFG
    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.cli.main import ray_train

    cfg_text = TAGGER_CFG
    for split in ("train", "dev"):
        cfg_text = cfg_text.replace(
            f"""[corpora.{split}]
@readers = "spacy-mi.SyntheticCorpus.v1\"""",
            f"""[corpora.{split}]
@readers = "spacy.Corpus.v1"
path = "{train_bin}\"""",
        )
    # strip the synthetic-reader kwargs that the DocBin reader doesn't take
    lines = [l for l in cfg_text.splitlines()
             if not l.startswith(("n_docs", "words_per_doc", "vocab_size",
                                  "n_tags", "seed", "world_seed", "shuffle"))]
    cfg = Config.from_str("\n".join(lines))
    cfg_path = tmp_path / "cfg.cfg"
    cfg_path.write_text(cfg.to_str())
    out = tmp_path / "model"
    rc = ray_train(
        Config.from_disk(cfg_path, overrides={"training.max_steps": 4,
                                              "training.eval_frequency": 2}),
        config_path=cfg_path, output_path=out, n_workers=1,
    )
    assert rc == 0
    assert (out / "model-last" / "config.cfg").exists()


def test_iob_to_biluo_always_valid():
    """Property: for ANY tag sequence, the output is structurally valid
    BILUO (B opens, I continues, L closes with matching labels; U/O stand
    alone) and marks the same token set as entity-covered for IOB2 input."""
    from hypothesis import given, strategies as st

    tag = st.one_of(
        st.just("O"),
        st.builds(lambda k, l: f"{k}-{l}",
                  st.sampled_from(["B", "I"]), st.sampled_from(["PER", "ORG", "X"])),
    )

    @given(st.lists(tag, max_size=12))
    def check(tags):
        out = iob_to_biluo(tags)
        assert len(out) == len(tags)
        open_label = None
        for t in out:
            kind, _, lab = t.partition("-")
            if kind == "B":
                assert open_label is None
                open_label = lab
            elif kind == "I":
                assert open_label == lab
            elif kind == "L":
                assert open_label == lab
                open_label = None
            else:
                assert t in ("O", f"U-{lab}")
                assert open_label is None
        assert open_label is None  # every span closed

    check()
