"""RQ1 / RQ3 / RQ4 aggregate tables (the reference's L4 layer).

Regenerates, from any 41-column taxonomy table:
  * RQ1: 19 strategies x 9 workflow stages, % of all labeled rows, plus the
    row-normalized variant        (ref RQs/RQ1/Results/RQ1_tests{,2}.csv)
  * RQ3: 21 properties x 9 repos %; strategies x repos %; and the
    strategy x property correlation matrix with the reference's
    "repo:(x%), ..." cell encoding  (ref RQs/RQ3/*.csv)
  * RQ4: test-method counts & percentages (ref RQs/RQ4/tests_methods.csv)

The reference ships only these OUTPUT tables (no analysis code exists in the
package — SURVEY.md §2); derivations here follow the documented schema and
are validated in tests/test_reference_golden.py against the shipped CSVs.
"""
from __future__ import annotations

from typing import Dict, List

import pandas as pd

from tosem2021_amd.analyze.taxonomy import (
    row_method, row_properties, row_stage, row_strategies)
from tosem2021_amd.extract.schema import (
    METHODS, PROPERTIES, REPOS, STAGES, STRATEGIES)


def rq1_strategies_by_stage(df: pd.DataFrame, normalize_rows: bool = False
                            ) -> pd.DataFrame:
    """19x9 matrix: percent of all labeled rows falling in (strategy, stage).

    normalize_rows=True adds the RQ1_tests2.csv-style 100%-per-row scaling.
    """
    stages = row_stage(df)
    strat_sets = row_strategies(df)
    counts = pd.DataFrame(0, index=STRATEGIES, columns=STAGES, dtype=float)
    for stage, strat in zip(stages, strat_sets):
        for s in strat:
            counts.loc[s, stage] += 1
    total = len(df)
    pct = counts / max(total, 1) * 100.0
    if normalize_rows:
        row_sums = counts.sum(axis=1).replace(0, 1)
        pct = counts.div(row_sums, axis=0) * 100.0
    pct.index.name = "Tests"
    return pct.round(4)


def rq3_properties_by_repo(df: pd.DataFrame) -> pd.DataFrame:
    """21 properties x 9 repos: % of the repo's rows testing each property."""
    props = row_properties(df)
    repos = df["Repo"].astype(str)
    counts = pd.DataFrame(0, index=PROPERTIES, columns=REPOS, dtype=float)
    repo_tot: Dict[str, int] = {r: 0 for r in REPOS}
    for repo, ps in zip(repos, props):
        if repo not in repo_tot:
            continue
        if ps:
            repo_tot[repo] += 1
        for p in ps:
            counts.loc[p, repo] += 1
    for r in REPOS:
        counts[r] = counts[r] / max(repo_tot[r], 1) * 100.0
    counts.index.name = "Property"
    return counts.round(4)


def rq3_strategies_by_repo(df: pd.DataFrame) -> pd.DataFrame:
    """19 strategies x 9 repos: % of the repo's strategy-labeled rows."""
    strat_sets = row_strategies(df)
    repos = df["Repo"].astype(str)
    counts = pd.DataFrame(0, index=STRATEGIES, columns=REPOS, dtype=float)
    repo_tot: Dict[str, int] = {r: 0 for r in REPOS}
    for repo, ss in zip(repos, strat_sets):
        if repo not in repo_tot:
            continue
        if ss:
            repo_tot[repo] += 1
        for s in ss:
            counts.loc[s, repo] += 1
    for r in REPOS:
        counts[r] = counts[r] / max(repo_tot[r], 1) * 100.0
    counts.index.name = "Tests"
    return counts.round(4)


def rq3_strategy_property_correlation(df: pd.DataFrame,
                                      encode_cells: str | bool = True
                                      ) -> pd.DataFrame:
    """Strategy x property matrix in the reference's three cell encodings
    (SURVEY.md §7 hard part 3):

      encode_cells=True / "percent": 'repo:(x%), ...'  (tests_correlate_rq3.csv)
      encode_cells="latex":          '$repo:x\%$, ...' (tests_correlate_rq4.csv)
      encode_cells=False / "count":  integer co-occurrence counts
                                     (tests_combined_correlate_rq3.csv)

    x% = share of that repo's strategy-labeled rows carrying both labels;
    a cell with no co-occurrence anywhere is the scalar 0.
    """
    if encode_cells == "count":
        encode_cells = False
    strat_sets = row_strategies(df)
    prop_sets = row_properties(df)
    repos = df["Repo"].astype(str)
    repo_tot: Dict[str, int] = {r: 0 for r in REPOS}
    co: Dict[tuple, Dict[str, int]] = {}
    for repo, ss, ps in zip(repos, strat_sets, prop_sets):
        if repo not in repo_tot:
            continue
        if ss:
            repo_tot[repo] += 1
        for s in ss:
            for p in ps:
                co.setdefault((s, p), {r: 0 for r in REPOS})[repo] += 1
    out = pd.DataFrame(index=STRATEGIES, columns=PROPERTIES, dtype=object)
    for s in STRATEGIES:
        for p in PROPERTIES:
            cell = co.get((s, p))
            if cell is None or not any(cell.values()):
                out.loc[s, p] = 0 if encode_cells else 0.0
                continue
            if encode_cells == "latex":
                out.loc[s, p] = "".join(
                    f"${r}:{round(cell[r] / max(repo_tot[r], 1) * 100, 2)}\\%$, "
                    for r in REPOS if cell[r])
            elif encode_cells:
                out.loc[s, p] = "".join(
                    f"{r}:({round(cell[r] / max(repo_tot[r], 1) * 100, 2)}%), "
                    for r in REPOS)
            else:
                out.loc[s, p] = sum(cell.values())
    out.index.name = "Tests"
    return out


def rq4_test_methods(df: pd.DataFrame) -> pd.DataFrame:
    """Method counts & percentages (tests_methods.csv schema), with the
    strategies and repos correlated to each method."""
    methods = row_method(df).to_numpy()
    strat_sets = row_strategies(df)
    repos = df["Repo"].astype(str).to_numpy()
    rows = []
    total = len(df)
    for m in METHODS:
        n = 0
        strategies_seen: List[str] = []
        repos_seen: List[str] = []
        n_correlate = 0
        for i in range(total):
            if methods[i] != m:
                continue
            n += 1
            ss = strat_sets[i]
            if ss:
                n_correlate += 1
            for s in ss:
                if s not in strategies_seen:
                    strategies_seen.append(s)
            r = repos[i]
            if r not in repos_seen and r in REPOS:
                repos_seen.append(r)
        rows.append({
            "Test_methods": m,
            "total_cases": n,
            "percentage": round(n / max(total, 1) * 100, 4),
            "correlate": n_correlate,
            "Strategy": ", ".join(strategies_seen),
            "Repos": ", ".join(repos_seen),
        })
    return pd.DataFrame(rows)


def write_all(df: pd.DataFrame, out_dir: str) -> Dict[str, str]:
    """Emit the full RQ table set under out_dir (mirrors RQs/ layout)."""
    import os

    paths = {}
    os.makedirs(os.path.join(out_dir, "RQ1", "Results"), exist_ok=True)
    os.makedirs(os.path.join(out_dir, "RQ3"), exist_ok=True)
    os.makedirs(os.path.join(out_dir, "RQ4"), exist_ok=True)
    t = rq1_strategies_by_stage(df)
    p = os.path.join(out_dir, "RQ1", "Results", "RQ1_tests.csv")
    t.to_csv(p); paths["rq1"] = p
    t2 = rq1_strategies_by_stage(df, normalize_rows=True)
    p = os.path.join(out_dir, "RQ1", "Results", "RQ1_tests2.csv")
    t2.to_csv(p); paths["rq1_norm"] = p
    pr = rq3_properties_by_repo(df)
    p = os.path.join(out_dir, "RQ3", "tests_prop_rq3.csv")
    pr.to_csv(p); paths["rq3_properties"] = p
    st = rq3_strategies_by_repo(df)
    p = os.path.join(out_dir, "RQ3", "tests_strategy_rq3.csv")
    st.to_csv(p); paths["rq3_strategies"] = p
    corr = rq3_strategy_property_correlation(df)
    p = os.path.join(out_dir, "RQ3", "tests_correlate_rq3.csv")
    corr.to_csv(p); paths["rq3_correlate"] = p
    cnt = rq3_strategy_property_correlation(df, encode_cells="count")
    p = os.path.join(out_dir, "RQ3", "tests_combined_correlate_rq3.csv")
    cnt.to_csv(p); paths["rq3_correlate_counts"] = p
    ltx = rq3_strategy_property_correlation(df, encode_cells="latex")
    p = os.path.join(out_dir, "RQ3", "tests_correlate_rq4.csv")
    ltx.to_csv(p); paths["rq3_correlate_latex"] = p
    st_t = st.transpose()
    p = os.path.join(out_dir, "RQ3", "tests_strategy_transpose_rq3.csv")
    st_t.to_csv(p); paths["rq3_strategies_transpose"] = p
    m = rq4_test_methods(df)
    p = os.path.join(out_dir, "RQ4", "tests_methods.csv")
    m.to_csv(p, index=False); paths["rq4"] = p
    return paths
