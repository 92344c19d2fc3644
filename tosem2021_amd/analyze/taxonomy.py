"""Taxonomy table loading + per-row label derivation.

Works on both the reference master dataset (RQs/taxonomy_test2.csv) and on
taxonomy CSVs produced by this framework's extract+classify pipeline — the
analysis layer (rq1/rq3/rq4) consumes the normalized form produced here.
"""
from __future__ import annotations

from typing import Dict, List, Set

import pandas as pd

from tosem2021_amd.extract.schema import (
    APPROX_TYPE_TO_STRATEGY, CHECKS_TYPE_TO_STRATEGY, ERROR_TYPE_TO_STRATEGY,
    PROPERTIES, STAGES, STRATEGIES, TAXONOMY_COLUMNS, canonical_property)

# Open-coding Category (69 observed values in the master CSV) -> RQ1's 9
# ML-workflow stages.  This mapping is re-derived (the reference ships no
# analysis scripts — SURVEY.md header), chosen to mirror the paper's stage
# semantics.
CATEGORY_TO_STAGE: Dict[str, str] = {
    # data acquisition / storage / IO
    "Data Input": "data_collection", "Data-Aquisition": "data_collection",
    "Data Storage": "data_collection", "Dataset": "data_collection",
    "IO": "data_collection", "File": "data_collection",
    "Data": "data_collection", "External": "data_collection",
    "Data Generation": "data_collection",
    # cleaning / preprocessing / schema
    "Data Cleaning": "data_cleaning", "Data Preprocessing": "data_cleaning",
    "Data Schema": "data_cleaning", "Data Migration": "data_cleaning",
    "Data Fusion": "data_cleaning", "Data-Fusion": "data_cleaning",
    "Type-checking": "data_cleaning",
    # labelling
    "Label": "data_labelling",
    # feature engineering
    "Feature Engineering": "feature_engin",
    "Feature Preprocessing": "feature_engin",
    "Feature preprocessing": "feature_engin",
    "Feature Processing": "feature_engin",
    "Feature Selection": "feature_engin",
    "Feature Importance": "feature_engin",
    # model training
    "Model": "model_training", "Model Fit": "model_training",
    "Model-Training": "model_training", "Training": "model_training",
    "Neural Network": "model_training", "Model Optimization": "model_training",
    "Optimization": "model_training", "Hyperparams": "model_training",
    "Model-Tuner": "model_training", "Model-Tunner": "model_training",
    "Model-Selection": "model_training", "Meta-Learning": "model_training",
    "Model Update": "model_training",
    "Information & Kalma Filter": "model_training",
    "Information & Kalman Filter": "model_training",
    # post-processing / analysis / evaluation
    "Model Postprocessing": "data_post", "Prediction": "data_post",
    "Evaluation": "data_post", "Data Analysis": "data_post",
    "Data-Analysis": "data_post", "Data Visualization": "data_post",
    "Object Detection": "data_post", "Object detection": "data_post",
    "Segmentation": "data_post", "Model Inspector": "data_post",
    "Anomaly": "data_post",
    # deployment / export / inference
    "Deployment": "model_deployment", "Model Export": "model_deployment",
    "Export Model": "model_deployment", "Inference": "model_deployment",
    "API": "model_deployment", "Network Utility": "model_deployment",
    # monitoring
    "Monitoring": "Monitoring", "Logging": "Monitoring",
    "Memory & Performance": "Monitoring", "Concurrency": "Monitoring",
    "Security": "Monitoring",
    # configuration / utility / test-infrastructure
    "Configuration": "config_utility", "Utility": "config_utility",
    "Dependency": "config_utility", "Decorator": "config_utility",
    "Mock": "config_utility", "Sanity": "config_utility",
    "Integration Test": "config_utility", "Integration-Test": "config_utility",
    "Regression Test": "config_utility",
}


def load_taxonomy(path: str) -> pd.DataFrame:
    """Load a 41-column taxonomy CSV (reference or regenerated)."""
    df = pd.read_csv(path)
    missing = [c for c in TAXONOMY_COLUMNS if c not in df.columns]
    if missing:
        raise ValueError(f"taxonomy CSV missing columns: {missing}")
    return df


def _flag(df: pd.DataFrame, col: str) -> pd.Series:
    return pd.to_numeric(df[col], errors="coerce").fillna(0).astype(int)


def _clean_str(s) -> str:
    if s is None or s != s:
        return ""
    return str(s).strip().rstrip("'")


def row_strategies(df: pd.DataFrame) -> List[Set[str]]:
    """Per-row strategy label sets (the same derivation TestCaseRow uses)."""
    status = _flag(df, "status_test")
    neg = _flag(df, "negative_test")
    log1 = _flag(df, "logical_statement")
    log2 = _flag(df, "logical_expression")
    null = _flag(df, "null_pointer")
    vrange = _flag(df, "value_range")
    errh = _flag(df, "error_handling")
    approx = _flag(df, "Approximation")
    out: List[Set[str]] = []
    for i in range(len(df)):
        s: Set[str] = set()
        if status.iat[i]:
            s.add("status_analysis")
        if neg.iat[i]:
            s.add("negative_test")
        if log1.iat[i] or log2.iat[i]:
            s.add("logical_condition")
        if null.iat[i]:
            s.add("Null_pointer")
        if vrange.iat[i]:
            s.add("value_range_analysis")
        if errh.iat[i]:
            et = ERROR_TYPE_TO_STRATEGY.get(
                _clean_str(df["Error_Type"].iat[i]).lower())
            if et:
                s.add(et)
        if approx.iat[i]:
            at = APPROX_TYPE_TO_STRATEGY.get(
                _clean_str(df["Approximation_Type"].iat[i]).lower())
            if at:
                s.add(at)
        ct = CHECKS_TYPE_TO_STRATEGY.get(_clean_str(df["checks_type"].iat[i]).lower())
        if ct:
            s.add(ct)
        out.append(s)
    return out


def row_properties(df: pd.DataFrame) -> List[Set[str]]:
    cols = [df[c] for c in ("Data", "Model", "Code", "Oracle")]
    out: List[Set[str]] = []
    for i in range(len(df)):
        s: Set[str] = set()
        for col in cols:
            c = canonical_property(col.iat[i])
            if c:
                s.add(c)
        out.append(s)
    return out


def row_method(df: pd.DataFrame) -> pd.Series:
    e2e = _flag(df, "end_to_end")
    integ = _flag(df, "Integration")
    reg = _flag(df, "regression")
    method = pd.Series("unit_test", index=df.index)
    method[reg > 0] = "regression"
    method[integ > 0] = "integration"
    method[e2e > 0] = "end_to_end"
    return method


def row_stage(df: pd.DataFrame,
              mapping: Dict[str, str] | None = None) -> pd.Series:
    """Workflow stage per row.

    Regenerated taxonomies already carry a stage name in Category; the
    reference master CSV carries open-coding categories that map through
    CATEGORY_TO_STAGE (default) or an explicit mapping (the RQ1-replication
    path passes stage_map.RQ1_RECOVERED_CATEGORY_TO_STAGE).
    """
    m = CATEGORY_TO_STAGE if mapping is None else mapping

    def to_stage(c) -> str:
        c = _clean_str(c)
        if c in STAGES:
            return c
        return m.get(c, "config_utility")

    return df["Category"].map(to_stage)
