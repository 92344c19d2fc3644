"""Unit tests: RQ pivots on a small hand-built taxonomy table."""
import os

import pandas as pd
import pytest

from tosem2021_amd.analyze.tables import (
    rq1_strategies_by_stage, rq3_properties_by_repo,
    rq3_strategy_property_correlation, rq4_test_methods, write_all)
from tosem2021_amd.analyze.taxonomy import load_taxonomy, row_method
from tosem2021_amd.extract.schema import (
    PROPERTIES, REPOS, STAGES, STRATEGIES, TAXONOMY_COLUMNS)


def _toy_df():
    rows = []
    base = {c: "" for c in TAXONOMY_COLUMNS}
    for c in ("regression", "Integration", "end_to_end", "status_test",
              "negative_test", "value_range", "null_pointer",
              "logical_statement", "logical_expression", "error_handling",
              "Approximation", "basic_comparizon"):
        base[c] = 0
    r1 = dict(base, Index=1, Labels="assertTrue(ok())", Category="Model",
              Repo="Ray", status_test=1, Model="Correctness")
    r2 = dict(base, Index=2, Labels="assertAlmostEqual(a,b)",
              Category="Data Preprocessing", Repo="Ray", Approximation=1,
              Approximation_Type="rounding_tolence", Data="Validity")
    r3 = dict(base, Index=3, Labels="raises ValueError",
              Category="Configuration", Repo="tpot", error_handling=1,
              Error_Type="ValueError", Integration=1)
    r4 = dict(base, Index=4, Labels="isinstance check", Category="Model",
              Repo="tpot", checks_type="instance_check", Model="Consistency")
    rows = [r1, r2, r3, r4]
    return pd.DataFrame(rows)[TAXONOMY_COLUMNS]


def test_rq1_pivot():
    t = rq1_strategies_by_stage(_toy_df())
    assert list(t.index) == STRATEGIES
    assert list(t.columns) == STAGES
    assert t.loc["status_analysis", "model_training"] == 25.0  # 1 of 4 rows
    assert t.loc["rounding_tolence", "data_cleaning"] == 25.0
    assert t.loc["value_error", "config_utility"] == 25.0


def test_rq3_properties_pivot():
    t = rq3_properties_by_repo(_toy_df())
    assert list(t.index) == PROPERTIES
    assert list(t.columns) == REPOS
    # Ray: 2 property-labeled rows, one Correctness, one Data Validity
    assert t.loc["Correctness", "Ray"] == 50.0
    assert t.loc["Data Validity", "Ray"] == 50.0
    assert t.loc["Consistency", "tpot"] == 100.0


def test_rq3_correlation_cells():
    t = rq3_strategy_property_correlation(_toy_df())
    cell = t.loc["status_analysis", "Correctness"]
    # Ray has 2 strategy-labeled rows; status_analysis+Correctness on 1 = 50%
    assert isinstance(cell, str) and "Ray:(50.0%)" in cell
    assert t.loc["memory_error", "Anomaly"] == 0


def test_rq4_methods():
    t = rq4_test_methods(_toy_df())
    d = t.set_index("Test_methods")
    assert d.loc["unit_test", "total_cases"] == 3
    assert d.loc["integration", "total_cases"] == 1
    assert abs(d.loc["unit_test", "percentage"] - 75.0) < 1e-6
    assert "value_error" in d.loc["integration", "Strategy"]
    assert "tpot" in d.loc["integration", "Repos"]


def test_write_all_and_figures(tmp_path):
    df = _toy_df()
    paths = write_all(df, str(tmp_path))
    for p in paths.values():
        assert os.path.exists(p)
    from tosem2021_amd.analyze.figures import write_figures
    figs = write_figures(df, str(tmp_path))
    for p in figs.values():
        assert os.path.exists(p)
        assert open(p).read().startswith("<svg")


def test_load_taxonomy_rejects_bad_schema(tmp_path):
    p = tmp_path / "bad.csv"
    pd.DataFrame({"a": [1]}).to_csv(p, index=False)
    with pytest.raises(ValueError):
        load_taxonomy(str(p))


def test_rq3_correlation_all_encodings():
    df = _toy_df()
    cnt = rq3_strategy_property_correlation(df, encode_cells="count")
    assert cnt.loc["status_analysis", "Correctness"] == 1
    ltx = rq3_strategy_property_correlation(df, encode_cells="latex")
    cell = ltx.loc["status_analysis", "Correctness"]
    assert cell.startswith("$Ray:") and "\\%$" in cell
