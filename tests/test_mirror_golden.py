"""Golden tests for the file-for-file RQ artifact mirror (analyze/mirror.py).

The reference ships 19 CSV artifacts + 6 plots under RQs/; write_mirror must
emit a counterpart for every one, with identical header/row vocabularies and
gated value agreement (Pearson).  Exact cell equality is impossible — the
shipped tables were computed on the study's unreleased per-test-case
intermediate (mirror.py docstring / scripts/rq_forensics.py) — so the gates
are schema-exact + correlation floors set just under measured values.
"""
import os

import pytest


@pytest.fixture(scope="module")
def mirror_result(tmp_path_factory):
    path = "/root/reference/RQs/taxonomy_test2.csv"
    if not os.path.exists(path):
        pytest.skip("reference corpus not mounted on this box")
    from tosem2021_amd.analyze.golden_mirror import mirror_diff
    from tosem2021_amd.analyze.mirror import write_mirror
    from tosem2021_amd.analyze.taxonomy import load_taxonomy
    out = str(tmp_path_factory.mktemp("mirror"))
    df = load_taxonomy(path)
    paths = write_mirror(df, out)
    return paths, mirror_diff(out, "/root/reference/RQs")


def test_every_shipped_csv_has_a_counterpart(mirror_result):
    paths, res = mirror_result
    assert res["n_files"] == 19
    for rel, v in res["files"].items():
        assert "missing" not in v, f"{rel}: {v}"


def test_schema_identity_all_files(mirror_result):
    """Header vocabulary and row-label sets identical for all 19 CSVs."""
    _, res = mirror_result
    bad = {rel: v for rel, v in res["files"].items() if not v.get("ok")}
    assert not bad, bad


def test_plot_counterparts_exist(mirror_result):
    """SVG counterparts for all 6 shipped plots (4 Rplots + 2 others)."""
    paths, _ = mirror_result
    for key in ("Rplot", "Rplot01", "Rplot03", "Rplot04",
                "fig_properties", "fig_strategies"):
        assert key in paths and os.path.getsize(paths[key]) > 500, key


# Correlation floors, set just under values measured on the reference master
# (round-2; see PROGRESS notes).  A regression in any derivation fails here.
GATES = {
    "RQ1/Results/RQ1_tests.csv": 0.70,       # >= 0.7
    "RQ1/Results/RQ1_tests2.csv": 0.85,
    "RQ3/properties_rq3.csv": 0.85,
    "RQ3/strategy_RQ3.csv": 0.85,
    "RQ3/tests_prop_rq3.csv": 0.80,
    "RQ3/tests_strategy_rq3.csv": 0.80,
    "RQ3/tests_strategy_rq32.csv": 0.80,
    "RQ3/tests_strategy_transpose_rq3.csv": 0.35,
    "RQ3/tests_combined_correlate.csv": 0.80,
    "RQ3/tests_combined_correlate_rq3.csv": 0.80,
    "RQ3/tests_correlate_rq3.csv": 0.60,
    "RQ3/tests_correlate_rq4.csv": 0.60,
}


def test_value_agreement_gates(mirror_result):
    _, res = mirror_result
    fails = {}
    for rel, floor in GATES.items():
        p = res["files"][rel].get("pearson")
        if p is None or p < floor:
            fails[rel] = (p, floor)
    assert not fails, fails


def test_rq1_exploits_recovered_stage_map(mirror_result):
    """RQ1 in the reference's true row semantics must
    correlate far better than the round-1 0.19."""
    _, res = mirror_result
    assert res["files"]["RQ1/Results/RQ1_tests.csv"]["pearson"] >= 0.70


def test_correlate_structure(mirror_result):
    """Nonzero-cell structure of the big correlate matrix agrees."""
    _, res = mirror_result
    assert res["files"]["RQ3/tests_correlate_rq3.csv"][
        "nonzero_jaccard"] >= 0.55


def test_mirror_runs_on_mined_style_taxonomy(tmp_path):
    """write_mirror must also work on taxonomies this framework mines
    (Category holds stage names directly; no reference dependency)."""
    import pandas as pd
    from tosem2021_amd.analyze.mirror import write_mirror
    from tosem2021_amd.extract.schema import TAXONOMY_COLUMNS
    rows = []
    for i, (repo, strat_flag) in enumerate(
            [("Apollo", "status_test"), ("Ray", "negative_test"),
             ("nni", "value_range"), ("tpot", "null_pointer")] * 3):
        row = {c: "" for c in TAXONOMY_COLUMNS}
        row.update({"Index": i, "Labels": f"assert x == {i}",
                    "Category": "model_training", "Cases": 1, "FileID": i,
                    "Repo": repo, "Data": "Consistency" if i % 2 else "",
                    strat_flag: 1})
        rows.append(row)
    df = pd.DataFrame(rows, columns=TAXONOMY_COLUMNS)
    paths = write_mirror(df, str(tmp_path))
    assert len(paths) >= 26
    for p in paths.values():
        assert os.path.getsize(p) > 0


def test_mirror_self_consistency(tmp_path):
    """golden_mirror comparing a mirror against ITSELF: every file ok with
    Pearson 1.0 — guards the comparator and the emitters jointly."""
    import pandas as pd

    from tosem2021_amd.analyze.golden_mirror import mirror_diff
    from tosem2021_amd.analyze.mirror import write_mirror
    from tosem2021_amd.extract.schema import TAXONOMY_COLUMNS
    rows = []
    for i in range(60):
        row = {c: "" for c in TAXONOMY_COLUMNS}
        row.update({"Index": i, "Labels": f"assertEqual(x, {i})",
                    "Category": "model_training", "Cases": 1, "FileID": i,
                    "Repo": ["Apollo", "Ray", "nni"][i % 3],
                    "Data": "Data Validity" if i % 2 else "",
                    "status_test": i % 2, "negative_test": (i + 1) % 2,
                    "error_handling": i % 3 == 0,
                    "Error_Type": "ValueError" if i % 3 == 0 else ""})
        rows.append(row)
    df = pd.DataFrame(rows, columns=TAXONOMY_COLUMNS)
    out = str(tmp_path / "m")
    write_mirror(df, out)
    res = mirror_diff(out, out)
    assert res["ok"], res
    for rel, v in res["files"].items():
        if "pearson" in v and v.get("cells", 0) > 2:
            # pearson degenerates to 0.0 when all compared cells are equal
            # (zero variance) — identical files may hit that legitimately
            assert v["pearson"] > 0.999 or \
                v.get("nonzero_jaccard", 1.0) == 1.0, (rel, v)
