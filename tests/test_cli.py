"""End-to-end CLI test over a synthetic mini corpus (no reference needed)."""
import json
import os
import subprocess
import sys
import textwrap

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, **kw):
    return subprocess.run([sys.executable, "-m", "tosem2021_amd.cli"] + args,
                          capture_output=True, text=True, cwd=REPO,
                          timeout=600, **kw)


@pytest.fixture
def mini_corpus(tmp_path):
    root = tmp_path / "corpus"
    proj = root / "tpot" / "v0.11.7" / "tests"
    proj.mkdir(parents=True)
    (proj / "test_pipeline.py").write_text(textwrap.dedent('''
        import pytest

        def test_fit_scores():
            model = object()
            score = 0.9
            assert score > 0.5
            self_check = isinstance(model, object)
            assert isinstance(model, object)

        def test_invalid_config():
            with pytest.raises(ValueError):
                raise ValueError("bad")
    '''))
    return str(root)


def test_mine_analyze_report_roundtrip(mini_corpus, tmp_path):
    tax = str(tmp_path / "tax.csv")
    r = _run(["mine", "--projects", "tpot", "--corpus-root", mini_corpus,
              "--out", tax])
    assert r.returncode == 0, r.stderr[-1500:]
    assert os.path.exists(tax)

    out = str(tmp_path / "rqs")
    r = _run(["analyze", "--taxonomy", tax, "--out", out])
    assert r.returncode == 0, r.stderr[-1500:]
    assert os.path.exists(os.path.join(out, "RQ4", "tests_methods.csv"))
    assert os.path.exists(os.path.join(out, "RQ3", "strategy_rq3.svg"))

    r = _run(["report", "--taxonomy", tax])
    assert r.returncode == 0
    assert "taxonomy rows" in r.stdout
    assert "value_error" in r.stdout


def test_train_then_classify_cli(mini_corpus, tmp_path):
    tax = str(tmp_path / "tax.csv")
    _run(["mine", "--projects", "tpot", "--corpus-root", mini_corpus,
          "--out", tax])
    ck = str(tmp_path / "ck")
    r = _run(["train", "--taxonomy", tax, "--model", "mltc-tiny",
              "--steps", "3", "--batch", "4", "--seq", "32",
              "--ckpt-dir", ck])
    assert r.returncode == 0, r.stderr[-1500:]
    res = json.loads(r.stdout[r.stdout.index("{"):])
    assert res["steps"] == 3

    out = str(tmp_path / "labeled.csv")
    r = _run(["classify", "--taxonomy", tax, "--ckpt-dir", ck,
              "--model", "mltc-tiny", "--seq", "32", "--out", out])
    assert r.returncode == 0, r.stderr[-1500:]
    from tosem2021_amd.analyze.taxonomy import load_taxonomy
    df = load_taxonomy(out)
    assert len(df) > 0


def test_cli_labels_command(tmp_path, capsys):
    """labels subcommand: L2 ingestion + lineage report."""
    import json
    import os

    import pytest

    if not os.path.isdir("/root/reference/selection/completed-labels"):
        pytest.skip("reference corpus not mounted on this box")
    from tosem2021_amd.cli import main
    out = str(tmp_path / "labels.json")
    main(["labels", "--taxonomy", "/root/reference/RQs/taxonomy_test2.csv",
          "--json", out])
    with open(out) as f:
        res = json.load(f)
    assert res["codebook"]["strategies_uncovered"] == []
    assert len(res["release_sheets"]) == 8
    assert all(v["coverage"] == 1.0 for v in res["lineage"].values())


def test_cli_golden_mirror(tmp_path):
    """analyze (mirror) + golden --mirror round trip on the reference."""
    import os

    import pytest

    if not os.path.exists("/root/reference/RQs/taxonomy_test2.csv"):
        pytest.skip("reference corpus not mounted on this box")
    from tosem2021_amd.cli import main
    out = str(tmp_path / "RQs")
    main(["analyze", "--taxonomy", "/root/reference/RQs/taxonomy_test2.csv",
          "--out", out])
    assert os.path.exists(os.path.join(out, "RQ3",
                                       "tests_correlate_assertion.csv"))
    with pytest.raises(SystemExit) as exc:
        main(["golden", "--ours", out, "--reference",
              "/root/reference/RQs", "--mirror"])
    assert exc.value.code == 0
