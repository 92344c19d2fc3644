#!/usr/bin/env python3
"""Forensics: pin the exact derivations behind the shipped RQ3/RQ1 tables.

Reads the reference master taxonomy and tests denominator hypotheses against
the shipped tests_strategy_rq32.csv raw block and RQ1_tests.csv cells.
Exploration tool (round-2); conclusions get encoded in analyze/mirror.py.
"""
from __future__ import annotations

import sys

import pandas as pd

sys.path.insert(0, ".")
from tosem2021_amd.analyze.taxonomy import load_taxonomy, row_strategies  # noqa
from tosem2021_amd.extract.schema import REPOS, STRATEGIES  # noqa

REF = "/root/reference/RQs"

df = load_taxonomy(f"{REF}/taxonomy_test2.csv")
print("rows:", len(df))
strats = row_strategies(df)
repos = df["Repo"].astype(str).str.strip()
print("repo values:", sorted(repos.unique()))

# counts per (repo, strategy)
cnt = {r: {s: 0 for s in STRATEGIES} for r in repos.unique()}
lab_rows = {r: 0 for r in repos.unique()}   # rows with >=1 strategy
tot_rows = {r: 0 for r in repos.unique()}
tot_labels = {r: 0 for r in repos.unique()}  # total strategy labels
for r, ss in zip(repos, strats):
    tot_rows[r] += 1
    if ss:
        lab_rows[r] += 1
    tot_labels[r] += len(ss)
    for s in ss:
        cnt[r][s] += 1

# shipped rq32 raw block
rq32 = pd.read_csv(f"{REF}/RQ3/tests_strategy_rq32.csv", encoding="utf-8-sig")
rq32 = rq32.set_index(rq32.columns[0])
raw = rq32.iloc[:19, :9].astype(float)
print("\nrq32 raw columns:", list(raw.columns))

ROWMAP = {  # rq32 row label -> our strategy label
    "status_analysis": "status_analysis", "value_error": "value_error",
    "runtime_error": "runtime_error", "memory_error": "memory_error",
    "type_error": "type_error", "import_error": "import_error",
    "key_error": "key_error", "AssertionError": "AssertionError",
    "FileError": "FileError", "NotImplementedError": "NotImplementedError",
    "negative_test": "negative_test", "logical_condition": "logical_condition",
    "Null_pointer": "Null_pointer", "value_range": "value_range_analysis",
    "absolute_relative_tolerence": "absolute_relative_tolerence",
    "error_bounding": "error_bounding", "rounding_tolence": "rounding_tolence",
    "instance_check": "instance_check", "sub_set_checks": "sub_set_checks",
}

for repo in raw.columns:
    print(f"\n== {repo}: rows={tot_rows.get(repo)} labeled={lab_rows.get(repo)}"
          f" labels={tot_labels.get(repo)}")
    for rowlab in list(raw.index[:19]):
        pct = raw.loc[rowlab, repo]
        if isinstance(pct, pd.Series):
            pct = pct.iloc[0]
        s = ROWMAP.get(str(rowlab))
        if s is None or repo not in cnt:
            continue
        n = cnt[repo][s]
        implied = n / pct * 100 if pct else None
        print(f"  {rowlab:30s} ship={pct:8.4f} ours_n={n:4d} "
              f"implied_denom={implied and round(implied, 1)}")
