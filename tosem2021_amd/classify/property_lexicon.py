"""Calibrated property lexicon: the rule lane's quality-property labeler.

The study's 21 quality-property labels (RQs/RQ3/tests_prop_rq3.csv:1) are
human judgments that raw assert text barely verbalizes — round 1's keyword
regexes scored micro-F1 0.068 against the gold rows.  This module replaces
them with a calibrated per-property lexicon: naive-Bayes log-odds weights
over cheap, fully-interpretable features (tokens + bigrams of the labeled
text and component, the rule engine's own predicted strategy/stage/
error-type labels, and the repo), with per-property decision thresholds
tuned for F1.

Calibration protocol (scripts/calibrate_property_lexicon.py): weights and
thresholds are fit on the EVEN-index half of the reference's 9,685 gold
rows and all reported scores are from the held-out odd half — the
committed artifact (artifacts/property_lexicon.json) never saw the rows it
is scored on.  Held-out micro-F1: 0.176 (round-1 rules: 0.068); the
per-property breakdown lives in artifacts/property_breakdown.json.
"""
from __future__ import annotations

import json
import math
import os
import re
from typing import Dict, List, Optional, Set

_TOK = re.compile(r"[A-Za-z_]+")

_DEFAULT_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                             "..", "..", "artifacts",
                             "property_lexicon.json")


def property_features(text: str, component: str = "", repo: str = "",
                      row=None) -> Set[str]:
    """Feature set for one taxonomy row.  `row` is an already-classified
    TestCaseRow (classify_text output); pass it to avoid re-classifying."""
    if row is None:
        from tosem2021_amd.classify.rules import classify_text
        row = classify_text(text, path=component)
    fs: Set[str] = set()
    for s in row.strategies():
        fs.add("S:" + s)
    fs.add("C:" + row.category)
    if repo:
        fs.add("R:" + repo)
    if row.error_type:
        fs.add("E:" + row.error_type)
    if row.approximation_type:
        fs.add("A:" + row.approximation_type)
    if row.checks_type:
        fs.add("K:" + row.checks_type)
    words: List[str] = []
    for w in _TOK.findall(f"{text} {component}".lower()):
        for p in w.split("_"):
            if 2 < len(p) < 25:
                words.append(p)
    fs.update("T:" + w for w in words)
    fs.update("B:" + a + "_" + b for a, b in zip(words, words[1:]))
    return fs


class PropertyLexicon:
    """Per-property log-odds weights + decision thresholds."""

    def __init__(self, classes: Dict[str, dict]):
        # classes: {property: {"prior": float, "threshold": float,
        #                      "weights": {feature: w}}}
        self.classes = classes

    def predict(self, feats: Set[str]) -> List[str]:
        out = []
        for prop, c in self.classes.items():
            s = c["prior"]
            w = c["weights"]
            for f in feats:
                s += w.get(f, 0.0)
            if s >= c["threshold"]:
                out.append((s - c["threshold"], prop))
        # strongest first (writes into the 4 Model/Data/Code/Oracle slots)
        return [p for _, p in sorted(out, reverse=True)]

    def save(self, path: str) -> None:
        with open(path, "w") as f:
            json.dump({"classes": self.classes}, f)

    @classmethod
    def load(cls, path: str) -> "PropertyLexicon":
        with open(path) as f:
            d = json.load(f)
        return cls(d["classes"])


_CACHED: Optional[PropertyLexicon] = None
_CACHED_MISSING = False


def default_lexicon() -> Optional[PropertyLexicon]:
    """The committed calibrated lexicon, or None if not present."""
    global _CACHED, _CACHED_MISSING
    if _CACHED is not None or _CACHED_MISSING:
        return _CACHED
    path = os.path.normpath(_DEFAULT_PATH)
    if not os.path.exists(path):
        _CACHED_MISSING = True
        return None
    _CACHED = PropertyLexicon.load(path)
    return _CACHED


def fit_lexicon(feat_rows: List[Set[str]], gold: List[Set[str]],
                properties: List[str], min_pos: int = 5,
                min_count: int = 2, max_features: int = 4000
                ) -> PropertyLexicon:
    """Fit weights + F1-optimal thresholds on the given (train) rows."""
    from collections import Counter
    classes: Dict[str, dict] = {}
    n = len(feat_rows)
    for prop in properties:
        pos = [i for i in range(n) if prop in gold[i]]
        neg = [i for i in range(n) if prop not in gold[i]]
        if len(pos) < min_pos:
            continue
        cpos: Counter = Counter()
        cneg: Counter = Counter()
        for i in pos:
            cpos.update(feat_rows[i])
        for i in neg:
            cneg.update(feat_rows[i])
        weights: Dict[str, float] = {}
        for f, c in cpos.items():
            if c < min_count:
                continue
            a = (c + 0.5) / (len(pos) + 1)
            b = (cneg[f] + 0.5) / (len(neg) + 1)
            weights[f] = math.log(a / b)
        if len(weights) > max_features:
            keep = sorted(weights.items(),
                          key=lambda kv: -abs(kv[1]) * cpos[kv[0]])
            weights = dict(keep[:max_features])
        # round before threshold tuning so stored and scored weights match
        weights = {k: round(v, 4) for k, v in weights.items()}
        prior = math.log(len(pos) / len(neg))

        def score(i: int) -> float:
            return prior + sum(weights.get(f, 0.0) for f in feat_rows[i])

        ranked = sorted(((score(i), prop in gold[i]) for i in range(n)),
                        key=lambda x: -x[0])
        best_f1, best_t = 0.0, float("inf")
        tp = fp = 0
        npos = len(pos)
        for s, g in ranked:
            if g:
                tp += 1
            else:
                fp += 1
            f1 = 2 * tp / max(2 * tp + fp + (npos - tp), 1)
            if f1 > best_f1:
                best_f1, best_t = f1, s
        classes[prop] = {"prior": prior, "threshold": best_t - 1e-9,
                         "weights": weights}
    return PropertyLexicon(classes)


def apply_to_row(row, text: str, component: str = "",
                 repo: str = "") -> None:
    """Overwrite a TestCaseRow's property slots with lexicon predictions
    (no-op when the committed artifact is absent)."""
    lex = default_lexicon()
    if lex is None:
        return
    feats = property_features(text, component, repo, row=row)
    props = lex.predict(feats)
    for slot, val in zip(("model", "data", "code", "oracle"), props):
        setattr(row, slot, val)
