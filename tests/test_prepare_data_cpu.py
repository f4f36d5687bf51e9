"""Data-prep tools: validate / split / token stats; model visualizer."""
import json
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO / "tools"))


def _write(tmp_path, lines):
    p = tmp_path / "data.jsonl"
    p.write_text("\n".join(lines) + "\n")
    return p


def test_validate_jsonl(tmp_path):
    import prepare_data

    good = _write(tmp_path, [json.dumps({"text": f"doc {i}"}) for i in range(5)])
    r = prepare_data.validate_jsonl(good)
    assert r["valid"] and r["ok"] == 5 and r["bad"] == 0

    bad = _write(tmp_path, ['{"text": "ok"}', "not json", '{"text": ""}'])
    r = prepare_data.validate_jsonl(bad)
    assert not r["valid"]
    assert r["ok"] == 1 and r["bad"] == 1 and r["empty"] == 1
    assert r["errors"][0]["line"] == 2


def test_split_jsonl(tmp_path):
    import prepare_data

    src = _write(tmp_path, [json.dumps({"text": f"d{i}"}) for i in range(100)])
    r = prepare_data.split_jsonl(src, tmp_path / "train.jsonl", tmp_path / "val.jsonl",
                                 val_fraction=0.1)
    assert r == {"total": 100, "train": 90, "val": 10}
    train = (tmp_path / "train.jsonl").read_text().splitlines()
    val = (tmp_path / "val.jsonl").read_text().splitlines()
    assert len(train) == 90 and len(val) == 10
    assert set(train).isdisjoint(set(val))


def test_token_stats(tmp_path):
    import prepare_data

    src = _write(tmp_path, [json.dumps({"text": "one two three"}),
                            json.dumps({"text": "four five"})])
    r = prepare_data.token_stats(src)
    assert r["docs"] == 2 and r["tokens"] == 5 and not r["exact"]
    assert r["max_tokens"] == 3


def test_visualize_model(capsys):
    import visualize_model

    visualize_model.main(["--config", str(REPO / "configs" / "model-config-sample.yaml")])
    out = capsys.readouterr().out
    assert "total params" in out
    assert "tok_embeddings.weight" in out
