"""Aux subsystems: tracer, metrics, YAML config."""
import json

import pytest

from tosem2021_amd.utils.config import (ExperimentConfig, apply_overrides,
                                        load_config)
from tosem2021_amd.utils.metrics import Metrics
from tosem2021_amd.utils.trace import Tracer


def test_tracer_chrome_format(tmp_path):
    tr = Tracer()
    with tr.trace("mine", project="ray"):
        with tr.trace("extract"):
            pass
    tr.instant("checkpoint_saved", step=3)
    p = tr.save(str(tmp_path / "trace.json"))
    data = json.load(open(p))
    evs = data["traceEvents"]
    assert len(evs) == 3
    names = {e["name"] for e in evs}
    assert names == {"mine", "extract", "checkpoint_saved"}
    x = next(e for e in evs if e["name"] == "mine")
    assert x["ph"] == "X" and x["dur"] > 0
    assert x["args"] == {"project": "ray"}


def test_metrics_registry(tmp_path):
    m = Metrics()
    m.inc("rows_extracted", 10, repo="ray")
    m.inc("rows_extracted", 5, repo="ray")
    m.set("loss", 1.5)
    m.observe_step(1, loss=1.5, lr=1e-4)
    snap = m.snapshot()
    assert snap["counters"]['rows_extracted{repo="ray"}'] == 15
    assert snap["gauges"]["loss"] == 1.5
    p = m.save_jsonl(str(tmp_path / "hist.jsonl"))
    rec = json.loads(open(p).read().strip())
    assert rec["step"] == 1 and rec["loss"] == 1.5
    text = m.prometheus_text()
    assert 'rows_extracted{repo="ray"} 15' in text


def test_yaml_config_roundtrip(tmp_path):
    p = tmp_path / "exp.yaml"
    p.write_text(
        "name: my-exp\n"
        "out_dir: results\n"
        "train:\n  model: mltc-tiny\n  lr: 0.001\n  bucket_mb: 32\n"
        "model_cfg:\n  n_layers: 3\n")
    cfg = load_config(str(p))
    assert cfg.name == "my-exp"
    assert cfg.train.model == "mltc-tiny"
    assert cfg.train.lr == 0.001
    assert cfg.train.bucket_mb == 32
    assert cfg.model_cfg.n_layers == 3
    # unknown keys rejected
    bad = tmp_path / "bad.yaml"
    bad.write_text("nonsense_key: 1\n")
    with pytest.raises(KeyError):
        load_config(str(bad))


def test_overrides():
    cfg = ExperimentConfig()
    apply_overrides(cfg, {"train.lr": 9.0, "name": "x"})
    assert cfg.train.lr == 9.0 and cfg.name == "x"
    with pytest.raises(KeyError):
        apply_overrides(cfg, {"train.nope": 1})
