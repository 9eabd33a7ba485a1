"""Logger output format (the W column IS the BASELINE metric definition:
cluster-scaled words per step — reference loggers.py:54 contract)."""
import json

from spacy_ray_amd.config.registry import registry


def _info(step, score=None):
    return {
        "epoch": 0, "step": step, "score": score,
        "other_scores": {"tag_acc": 0.5, "speed": 1000.0} if score is not None else {},
        "losses": {"tagger": 1.25}, "words": 640, "words_scaled": 1280,
        "words_seen": 640 * (step + 1), "checkpoints": [],
    }


def test_console_logger_table(capsys):
    registry.ensure_populated()
    setup = registry.loggers.get("spacy-ray.ConsoleLogger.v1")()
    print_row, finalize = setup(None)
    print_row(_info(10, score=0.5))
    print_row(_info(20, score=0.6))
    finalize()
    out = capsys.readouterr().out
    lines = [l for l in out.splitlines() if l.strip()]
    assert "Loss tagger" in lines[0] and "Score" in lines[0]
    assert "tag_acc" in lines[0]
    assert "speed" not in lines[0]  # excluded from score columns
    # cluster-scaled words column
    assert "1280" in lines[1]
    assert len(lines) == 3


def test_jsonl_logger_writes_records(tmp_path):
    registry.ensure_populated()
    path = tmp_path / "log.jsonl"
    setup = registry.loggers.get("spacy-mi.JsonlLogger.v1")(path=str(path), console=False)
    print_row, finalize = setup(None)
    print_row(_info(5, score=0.4))
    finalize()
    rec = json.loads(path.read_text().strip())
    assert rec["step"] == 5 and rec["losses"]["tagger"] == 1.25
