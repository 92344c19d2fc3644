#!/usr/bin/env python3
"""Calibrate the property lexicon AND the strategy stack on the gold rows.

Protocol (property_lexicon.py / strategy_stack.py docstrings): fit on
even-index rows, score on the held-out odd half, write:
  artifacts/property_lexicon.json    — property lexicon (train half)
  artifacts/property_breakdown.json  — held-out per-property P/R/F1
  artifacts/strategy_stack.json      — strategy lexicon + per-class
                                       rules/lexicon/union/intersect modes
Run: python scripts/calibrate_property_lexicon.py
"""
from __future__ import annotations

import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from tosem2021_amd.analyze.taxonomy import (  # noqa: E402
    load_taxonomy, row_properties)
from tosem2021_amd.classify.property_lexicon import (  # noqa: E402
    fit_lexicon, property_features)
from tosem2021_amd.extract.schema import PROPERTIES  # noqa: E402

ROOT = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..")


def main() -> None:
    from tosem2021_amd.analyze.taxonomy import row_strategies
    from tosem2021_amd.classify.rules import classify_text
    from tosem2021_amd.classify.strategy_stack import StrategyStack
    from tosem2021_amd.extract.schema import STRATEGIES

    df = load_taxonomy("/root/reference/RQs/taxonomy_test2.csv")
    gold = row_properties(df)
    gold_s = row_strategies(df)
    texts = df["Labels"].astype(str).tolist()
    comps = df["Component"].astype(str).tolist()
    repos = df["Repo"].astype(str).tolist()
    print("extracting features for", len(df), "rows ...")
    rule_pred = []
    feats = []
    for i in range(len(df)):
        row = classify_text(texts[i], path=comps[i])
        rule_pred.append(set(row.strategies()))
        feats.append(property_features(texts[i], comps[i], repos[i],
                                       row=row))
    train = [i for i in range(len(df)) if i % 2 == 0]
    test = [i for i in range(len(df)) if i % 2 == 1]

    # ---- strategy stack ----
    slex = fit_lexicon([feats[i] for i in train],
                       [gold_s[i] for i in train], STRATEGIES)
    lex_pred = [set(slex.predict(feats[i])) for i in range(len(df))]

    def class_f1(pred_fn, s, idx):
        tp = fp = fn = 0
        for i in idx:
            a, b = pred_fn(i, s), s in gold_s[i]
            if a and b:
                tp += 1
            elif a:
                fp += 1
            elif b:
                fn += 1
        return 2 * tp / max(2 * tp + fp + fn, 1)

    modes_fns = {
        "rules": lambda i, s: s in rule_pred[i],
        "lexicon": lambda i, s: s in lex_pred[i],
        "union": lambda i, s: s in rule_pred[i] or s in lex_pred[i],
        "intersect": lambda i, s: s in rule_pred[i] and s in lex_pred[i],
    }
    modes = {s: max(modes_fns, key=lambda m: class_f1(modes_fns[m], s, train))
             for s in STRATEGIES}
    stack = StrategyStack(slex, modes)
    tp = fp = fn = 0
    for i in test:
        pred = stack.predict(feats[i], rule_pred[i])
        g = gold_s[i]
        tp += len(pred & g)
        fp += len(pred - g)
        fn += len(g - pred)
    sp = tp / max(tp + fp, 1)
    sr = tp / max(tp + fn, 1)
    sf1 = 2 * sp * sr / max(sp + sr, 1e-9)
    stack.save(os.path.join(ROOT, "artifacts", "strategy_stack.json"))
    print(json.dumps({"strategy_stack_heldout":
                      {"precision": round(sp, 4), "recall": round(sr, 4),
                       "f1": round(sf1, 4)}, "modes": modes}))
    lex = fit_lexicon([feats[i] for i in train], [gold[i] for i in train],
                      PROPERTIES)
    # held-out scoring
    per = {p: {"tp": 0, "fp": 0, "fn": 0} for p in PROPERTIES}
    tp = fp = fn = 0
    for i in test:
        pred = set(lex.predict(feats[i]))
        g = gold[i]
        tp += len(pred & g)
        fp += len(pred - g)
        fn += len(g - pred)
        for p in pred | g:
            if p in pred and p in g:
                per[p]["tp"] += 1
            elif p in pred:
                per[p]["fp"] += 1
            else:
                per[p]["fn"] += 1
    prec = tp / (tp + fp) if tp + fp else 0.0
    rec = tp / (tp + fn) if tp + fn else 0.0
    f1 = 2 * prec * rec / (prec + rec) if prec + rec else 0.0
    breakdown = {}
    for p, d in per.items():
        pp = d["tp"] / (d["tp"] + d["fp"]) if d["tp"] + d["fp"] else 0.0
        rr = d["tp"] / (d["tp"] + d["fn"]) if d["tp"] + d["fn"] else 0.0
        breakdown[p] = {
            "precision": round(pp, 4), "recall": round(rr, 4),
            "f1": round(2 * pp * rr / (pp + rr), 4) if pp + rr else 0.0,
            "support": d["tp"] + d["fn"],
        }
    out = {
        "protocol": "fit on even-index gold rows, scored on held-out odd",
        "heldout_micro": {"precision": round(prec, 4),
                          "recall": round(rec, 4), "f1": round(f1, 4)},
        "per_property": breakdown,
    }
    lex.save(os.path.join(ROOT, "artifacts", "property_lexicon.json"))
    with open(os.path.join(ROOT, "artifacts",
                           "property_breakdown.json"), "w") as f:
        json.dump(out, f, indent=1)
    print(json.dumps(out["heldout_micro"]))
    print("wrote artifacts/property_lexicon.json + property_breakdown.json")


if __name__ == "__main__":
    main()
