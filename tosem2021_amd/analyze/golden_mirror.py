"""Golden diff for the file-for-file RQ mirror (analyze/mirror.py).

For every CSV the reference ships under RQs/, checks against our mirrored
counterpart:
  * schema identity — identical header vocabulary and row-label sets
    (exact; spelling variants included);
  * value agreement — Pearson correlation between numeric cells.  Exact
    cell equality is impossible: the shipped tables were computed on the
    study's unreleased per-test-case intermediate (mirror.py docstring), so
    correlation over the released per-assertion master is the strongest
    honest check.
  * encoded tables ('repo:(x%)' / '$repo:x\\%$' cells) are parsed back into
    (row, property, repo) -> pct triples and compared by Pearson plus
    nonzero-structure Jaccard.
"""
from __future__ import annotations

import csv
import math
import os
import re
from typing import Dict, List, Optional, Tuple


def _pearson(a: List[float], b: List[float]) -> float:
    n = len(a)
    if n < 2:
        return 0.0
    ma = sum(a) / n
    mb = sum(b) / n
    cov = sum((x - ma) * (y - mb) for x, y in zip(a, b))
    va = math.sqrt(sum((x - ma) ** 2 for x in a))
    vb = math.sqrt(sum((y - mb) ** 2 for y in b))
    return cov / (va * vb) if va and vb else 0.0


def _read_rows(path: str) -> Optional[List[List[str]]]:
    if not os.path.exists(path):
        return None
    with open(path, newline="", encoding="utf-8-sig") as f:
        return [row for row in csv.reader(f)]


def _num(s: str) -> Optional[float]:
    try:
        return float(s)
    except (TypeError, ValueError):
        return None


def _table_cells(rows: List[List[str]]) -> Dict[Tuple[str, int], float]:
    """Numeric cells keyed by (row label, column position)."""
    out: Dict[Tuple[str, int], float] = {}
    for r in rows[1:]:
        if not r or not r[0]:
            continue
        for j, cell in enumerate(r[1:]):
            v = _num(cell)
            if v is not None:
                out[(r[0], j)] = v
    return out


def compare_numeric(ours_csv: str, ref_csv: str) -> dict:
    ours, ref = _read_rows(ours_csv), _read_rows(ref_csv)
    if ours is None or ref is None:
        return {"ok": False, "reason": "missing file",
                "missing": ours_csv if ours is None else ref_csv}
    header_ok = [h.strip() for h in ours[0]] == [h.strip() for h in ref[0]]
    our_labels = {r[0] for r in ours[1:] if r and r[0]}
    ref_labels = {r[0] for r in ref[1:] if r and r[0]}
    labels_ok = our_labels == ref_labels
    oc, rc = _table_cells(ours), _table_cells(ref)
    keys = sorted(set(oc) & set(rc))
    corr = _pearson([oc[k] for k in keys], [rc[k] for k in keys])
    return {"ok": bool(header_ok and labels_ok), "header_ok": bool(header_ok),
            "labels_ok": bool(labels_ok), "cells": len(keys),
            "pearson": round(corr, 4)}


_CELL_RE = re.compile(r"\$?([A-Za-z_0-9-]+):\(?([0-9.]+)\\?%\)?\$?")


def _encoded_cells(rows: List[List[str]]
                   ) -> Dict[Tuple[str, int, str], float]:
    out: Dict[Tuple[str, int, str], float] = {}
    for r in rows[1:]:
        if not r or not r[0]:
            continue
        for j, cell in enumerate(r[1:]):
            v = _num(cell)
            if v is not None:
                continue        # scalar 0 handled via absence
            for repo, pct in _CELL_RE.findall(cell or ""):
                out[(r[0], j, repo)] = float(pct)
    return out


def compare_encoded(ours_csv: str, ref_csv: str) -> dict:
    ours, ref = _read_rows(ours_csv), _read_rows(ref_csv)
    if ours is None or ref is None:
        return {"ok": False, "reason": "missing file",
                "missing": ours_csv if ours is None else ref_csv}
    header_ok = [h.strip() for h in ours[0]] == [h.strip() for h in ref[0]]
    our_labels = {r[0] for r in ours[1:] if r and r[0]}
    ref_labels = {r[0] for r in ref[1:] if r and r[0]}
    labels_ok = our_labels == ref_labels
    oc, rc = _encoded_cells(ours), _encoded_cells(ref)
    on = {k for k, v in oc.items() if v > 0}
    rn = {k for k, v in rc.items() if v > 0}
    union = on | rn
    jacc = len(on & rn) / len(union) if union else 1.0
    keys = sorted(union)
    corr = _pearson([oc.get(k, 0.0) for k in keys],
                    [rc.get(k, 0.0) for k in keys])
    return {"ok": bool(header_ok and labels_ok), "header_ok": bool(header_ok),
            "labels_ok": bool(labels_ok), "nonzero_jaccard": round(jacc, 4),
            "pearson": round(corr, 4), "cells": len(keys)}


def compare_rq4_methods(ours_csv: str, ref_csv: str) -> dict:
    ours, ref = _read_rows(ours_csv), _read_rows(ref_csv)
    if ours is None or ref is None:
        return {"ok": False, "reason": "missing file"}
    header_ok = [h.strip() for h in ours[0]] == [h.strip() for h in ref[0]]
    our_m = [r[0] for r in ours[1:] if r and r[0]]
    ref_m = [r[0] for r in ref[1:] if r and r[0]]
    methods_ok = our_m == ref_m
    dominant_ok = False
    best, best_n = None, -1.0
    for r in ours[1:]:
        if r and r[0] and _num(r[1]) is not None and _num(r[1]) > best_n:
            best, best_n = r[0], _num(r[1])
    dominant_ok = best == "unit_test"
    return {"ok": bool(header_ok and methods_ok and dominant_ok),
            "header_ok": bool(header_ok), "methods_ok": bool(methods_ok),
            "unit_test_dominant": bool(dominant_ok)}


# (shipped filename, comparison kind) — the complete RQs/ CSV inventory.
MIRROR_FILES: List[Tuple[str, str]] = [
    ("RQ1/Results/RQ1_tests.csv", "numeric"),
    ("RQ1/Results/RQ1_tests2.csv", "numeric"),
    ("RQ3/properties_rq3.csv", "numeric"),
    ("RQ3/strategy_RQ3.csv", "numeric"),
    ("RQ3/tests_prop_rq3.csv", "numeric"),
    ("RQ3/tests_strategy_rq3.csv", "numeric"),
    ("RQ3/tests_strategy_rq32.csv", "numeric"),
    ("RQ3/tests_strategy_transpose_rq3.csv", "numeric"),
    ("RQ3/tests_combined_correlate.csv", "numeric"),
    ("RQ3/tests_combined_correlate_rq3.csv", "numeric"),
    ("RQ3/tests_correlate_rq3.csv", "encoded"),
    ("RQ3/tests_correlate_rq4.csv", "encoded"),
    ("RQ3/tests_correlate_assertion.csv", "encoded"),
    ("RQ3/tests_correlate_FileError.csv", "encoded"),
    ("RQ3/tests_correlate_RuntimeError.csv", "encoded"),
    ("RQ3/tests_correlate_logical.csv", "encoded"),
    ("RQ4/tests_methods.csv", "rq4"),
    ("RQ4/tests_methods_v2.csv", "rq4"),
    ("RQ4/tests_methods_v3.csv", "rq4"),
]


def mirror_diff(ours_dir: str, ref_dir: str) -> dict:
    res: Dict[str, dict] = {}
    for rel, kind in MIRROR_FILES:
        ours, ref = os.path.join(ours_dir, rel), os.path.join(ref_dir, rel)
        if kind == "numeric":
            res[rel] = compare_numeric(ours, ref)
        elif kind == "encoded":
            res[rel] = compare_encoded(ours, ref)
        else:
            res[rel] = compare_rq4_methods(ours, ref)
    res_all = {"files": res,
               "n_files": len(MIRROR_FILES),
               "schema_ok": all(v.get("ok") for v in res.values()),
               "ok": all(v.get("ok") for v in res.values())}
    return res_all
