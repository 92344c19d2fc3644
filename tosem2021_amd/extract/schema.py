"""The taxonomy row schema and label vocabularies of the study.

Mirrors the reference's master dataset RQs/taxonomy_test2.csv:1 (41 columns,
9,685 logical rows) and the aggregate vocabularies:
  * 19 test strategies   — RQs/RQ1/Results/RQ1_tests.csv rows 2-20
  * 21 quality properties — RQs/RQ3/tests_prop_rq3.csv:1 columns
  * 9 ML workflow stages — RQs/RQ1/Results/RQ1_tests.csv:1 columns
  * 4 test methods        — RQs/RQ4/tests_methods.csv:2-5
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

# The reference master-CSV column order (taxonomy_test2.csv:1).
TAXONOMY_COLUMNS: List[str] = [
    "Index", "Labels", "Category", "Category2", "Cases", "FileID", "Component",
    "Repo", "Data", "Model", "Code", "Oracle", "Data_Generation", "regression",
    "Integration", "end_to_end", "boundary", "sanity", "mock_test",
    "Pseaudo_Oracle", "periodic_validation", "example_test",
    "static_inspection_test", "Experimental_benchmark_test", "roboustness",
    "blob_performance", "API", "ThreadTest", "Approximation",
    "Approximation_Type", "error_handling", "Error_Type", "value_range",
    "logical_statement", "logical_expression", "null_pointer", "status_test",
    "negative_test", "checks_type", "basic_comparizon", "None_above",
]

# 19 strategies — RQ1_tests.csv row labels, in file order.
STRATEGIES: List[str] = [
    "status_analysis", "negative_test", "logical_condition", "value_error",
    "runtime_error", "memory_error", "type_error", "import_error", "key_error",
    "AssertionError", "FileError", "NotImplementedError", "Null_pointer",
    "value_range_analysis", "absolute_relative_tolerence", "error_bounding",
    "rounding_tolence", "instance_check", "sub_set_checks",
]

# 21 canonical quality properties — tests_prop_rq3.csv:1 columns.
PROPERTIES: List[str] = [
    "Consistency", "Data Distribution", "Data Validity", "Completeness",
    "Correctness", "Robustness", "Efficiency", "Data Relation", "Scalability",
    "Features Importance", "Data Restoration and Recoverability",
    "Concurrency and Parallelism", "Uncertainty", "Anomaly",
    "Data Migration Loss and Corruption", "Bias and Fairness",
    "Security and Privacy", "Data Uniqueness", "Data Timeliness",
    "Data Integration Integrity", "Compatibility and Portability",
]

# 9 ML workflow stages — RQ1_tests.csv:1 column labels.
STAGES: List[str] = [
    "data_collection", "data_cleaning", "data_labelling", "feature_engin",
    "model_training", "data_post", "model_deployment", "Monitoring",
    "config_utility",
]

# 4 test methods — tests_methods.csv:2-5.
METHODS: List[str] = ["unit_test", "regression", "integration", "end_to_end"]

# The nine subject repos, taxonomy spelling (row counts per SURVEY.md §1 L3).
REPOS: List[str] = [
    "Apollo", "Nupic", "DeepSpeech2", "Ray", "auto_sklearn", "autokeras",
    "tpot", "nni", "google_automl",
]
REFERENCE_ROW_COUNTS: Dict[str, int] = {
    "Apollo": 3479, "Nupic": 1793, "DeepSpeech2": 1139, "Ray": 725,
    "auto_sklearn": 640, "autokeras": 574, "tpot": 511, "nni": 500,
    "google_automl": 324,
}

# ---------------------------------------------------------------------------
# Canonicalization of the free-text property labels used in the master CSV's
# Data/Model/Code/Oracle columns to the 21 RQ3 property names.  The master
# dataset holds ~50 raw open-coding values (e.g. "Roboustness",
# "Relation & Association"); the RQ3 tables aggregate them.
PROPERTY_CANON: Dict[str, str] = {
    "consistency": "Consistency",
    "distribution": "Data Distribution",
    "data distribution": "Data Distribution",
    "validity": "Data Validity",
    "data validity": "Data Validity",
    "completeness": "Completeness",
    "correctness": "Correctness",
    "robustness": "Robustness",
    "roboustness": "Robustness",
    "fault tolerance": "Robustness",
    "efficiency": "Efficiency",
    "computing efficiency": "Efficiency",
    "training efficiency": "Efficiency",
    "resource usage": "Efficiency",
    "time behaviour": "Efficiency",
    "memory allocation": "Efficiency",
    "relation & association": "Data Relation",
    "data relation": "Data Relation",
    "entities-relation oracle": "Data Relation",
    "scalability": "Scalability",
    "statistical evidence/ explainability": "Features Importance",
    "feature importance": "Features Importance",
    "features importance": "Features Importance",
    "recoverability": "Data Restoration and Recoverability",
    "restoration": "Data Restoration and Recoverability",
    "concurrency": "Concurrency and Parallelism",
    "parallelism": "Concurrency and Parallelism",
    "uncertainty": "Uncertainty",
    "anomaly": "Anomaly",
    "data error": "Anomaly",
    "migration": "Data Migration Loss and Corruption",
    "data loss": "Data Migration Loss and Corruption",
    "bias": "Bias and Fairness",
    "fairness": "Bias and Fairness",
    "security": "Security and Privacy",
    "privacy": "Security and Privacy",
    "uniqueness": "Data Uniqueness",
    "data uniqueness": "Data Uniqueness",
    "timeliness": "Data Timeliness",
    "data timeliness": "Data Timeliness",
    "integration integrity": "Data Integration Integrity",
    "compatibility": "Compatibility and Portability",
    "portability": "Compatibility and Portability",
}


def canonical_property(raw: Optional[str]) -> Optional[str]:
    if raw is None or raw != raw:  # None / NaN
        return None
    key = str(raw).strip().lower()
    if key in PROPERTY_CANON:
        return PROPERTY_CANON[key]
    for frag, canon in PROPERTY_CANON.items():
        if frag in key:
            return canon
    return None


# Error_Type values -> strategy names (master CSV Error_Type column).
ERROR_TYPE_TO_STRATEGY: Dict[str, str] = {
    "valueerror": "value_error",
    "runtimeerror": "runtime_error",
    "memoryerror": "memory_error",
    "typeerror": "type_error",
    "importerror": "import_error",
    "keyerror": "key_error",
    "assertionerror": "AssertionError",
    "fileerror": "FileError",
    "filenotfounderror": "FileError",
    "notimplementederror": "NotImplementedError",
    "nullptr": "Null_pointer",
    "nullpointer": "Null_pointer",
}

CHECKS_TYPE_TO_STRATEGY: Dict[str, str] = {
    "instance_check": "instance_check",
    "sub_set_checks": "sub_set_checks",
    "greater_checks": "value_range_analysis",
}

APPROX_TYPE_TO_STRATEGY: Dict[str, str] = {
    "absolute_relative_tolerence": "absolute_relative_tolerence",
    "rounding_tolence": "rounding_tolence",
    "error_bounding": "error_bounding",
}


# Inverse mapping: strategy label -> the taxonomy columns that encode it
# (used when writing model-predicted labels back into the 41-column schema).
STRATEGY_TO_COLUMNS = {
    "status_analysis": {"flags": {"status_test": 1}},
    "negative_test": {"flags": {"negative_test": 1}},
    "logical_condition": {"flags": {"logical_expression": 1}},
    "Null_pointer": {"flags": {"null_pointer": 1}},
    "value_range_analysis": {"flags": {"value_range": 1}},
    "value_error": {"flags": {"error_handling": 1}, "error_type": "ValueError"},
    "runtime_error": {"flags": {"error_handling": 1}, "error_type": "RuntimeError"},
    "memory_error": {"flags": {"error_handling": 1}, "error_type": "MemoryError"},
    "type_error": {"flags": {"error_handling": 1}, "error_type": "TypeError"},
    "import_error": {"flags": {"error_handling": 1}, "error_type": "ImportError"},
    "key_error": {"flags": {"error_handling": 1}, "error_type": "KeyError"},
    "AssertionError": {"flags": {"error_handling": 1},
                       "error_type": "AssertionError"},
    "FileError": {"flags": {"error_handling": 1}, "error_type": "FileError"},
    "NotImplementedError": {"flags": {"error_handling": 1},
                            "error_type": "NotImplementedError"},
    "absolute_relative_tolerence": {
        "flags": {"Approximation": 1},
        "approximation_type": "absolute_relative_tolerence"},
    "rounding_tolence": {"flags": {"Approximation": 1},
                         "approximation_type": "rounding_tolence"},
    "error_bounding": {"flags": {"Approximation": 1},
                       "approximation_type": "error_bounding"},
    "instance_check": {"checks_type": "instance_check"},
    "sub_set_checks": {"checks_type": "sub_set_checks"},
}


@dataclass
class TestCaseRow:
    """One extracted/labeled test case or assertion (one taxonomy row)."""

    index: int = 0
    labels: str = ""                  # free-text description / assert source
    category: str = ""                # ML-pipeline category (open coding)
    category2: str = ""
    cases: int = 1
    file_id: int = 0
    component: str = ""
    repo: str = ""
    # tested-property columns (canonical property names or "")
    data: str = ""
    model: str = ""
    code: str = ""
    oracle: str = ""
    data_generation: str = ""
    # binary strategy/kind flags (0/1)
    flags: Dict[str, int] = field(default_factory=dict)
    approximation_type: str = ""
    error_type: str = ""
    checks_type: str = ""

    BINARY_FLAGS = [
        "regression", "Integration", "end_to_end", "boundary", "sanity",
        "mock_test", "Pseaudo_Oracle", "periodic_validation", "example_test",
        "static_inspection_test", "Experimental_benchmark_test", "roboustness",
        "blob_performance", "API", "ThreadTest", "Approximation",
        "error_handling", "value_range", "logical_statement",
        "logical_expression", "null_pointer", "status_test", "negative_test",
        "basic_comparizon", "None_above",
    ]

    def to_csv_row(self) -> List[str]:
        f = self.flags
        return [
            str(self.index), self.labels, self.category, self.category2,
            str(self.cases), str(self.file_id), self.component, self.repo,
            self.data, self.model, self.code, self.oracle,
            self.data_generation,
            *(str(f.get(k, 0)) for k in ("regression", "Integration",
                                         "end_to_end", "boundary", "sanity",
                                         "mock_test", "Pseaudo_Oracle",
                                         "periodic_validation", "example_test",
                                         "static_inspection_test",
                                         "Experimental_benchmark_test",
                                         "roboustness", "blob_performance",
                                         "API", "ThreadTest", "Approximation")),
            self.approximation_type,
            str(f.get("error_handling", 0)), self.error_type,
            *(str(f.get(k, 0)) for k in ("value_range", "logical_statement",
                                         "logical_expression", "null_pointer",
                                         "status_test", "negative_test")),
            self.checks_type,
            str(f.get("basic_comparizon", 0)), str(f.get("None_above", 0)),
        ]

    @property
    def method(self) -> str:
        """unit_test | regression | integration | end_to_end (RQ4)."""
        if self.flags.get("end_to_end"):
            return "end_to_end"
        if self.flags.get("Integration"):
            return "integration"
        if self.flags.get("regression"):
            return "regression"
        return "unit_test"

    def strategies(self) -> List[str]:
        """The RQ1/RQ3 strategy labels this row carries."""
        out = []
        if self.flags.get("status_test"):
            out.append("status_analysis")
        if self.flags.get("negative_test"):
            out.append("negative_test")
        if self.flags.get("logical_statement") or self.flags.get("logical_expression"):
            out.append("logical_condition")
        if self.flags.get("null_pointer"):
            out.append("Null_pointer")
        if self.flags.get("value_range"):
            out.append("value_range_analysis")
        et = ERROR_TYPE_TO_STRATEGY.get(self.error_type.strip().lower().rstrip("'"))
        if self.flags.get("error_handling") and et:
            out.append(et)
        at = APPROX_TYPE_TO_STRATEGY.get(
            self.approximation_type.strip().lower().rstrip("'"))
        if self.flags.get("Approximation") and at:
            out.append(at)
        ct = CHECKS_TYPE_TO_STRATEGY.get(self.checks_type.strip().lower().rstrip("'"))
        if ct:
            out.append(ct)
        return sorted(set(out), key=STRATEGIES.index)

    def properties(self) -> List[str]:
        out = []
        for raw in (self.data, self.model, self.code, self.oracle):
            c = canonical_property(raw)
            if c:
                out.append(c)
        return sorted(set(out), key=PROPERTIES.index)
