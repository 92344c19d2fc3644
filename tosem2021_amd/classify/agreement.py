"""Agreement of the rule classifier against the study's human labels.

The replication metric for the open-coding step: run classify.rules over the
raw labeled text (the `Labels` column) of the reference master dataset and
score the predicted strategy/method/property labels against the study's own
columns.  (Exact replication of human judgment calls is the stated research
risk — SURVEY.md §7 hard part 1; this module quantifies it.)
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Set

import pandas as pd

from tosem2021_amd.analyze.taxonomy import (
    row_method, row_properties, row_strategies)
from tosem2021_amd.classify.rules import classify_text
from tosem2021_amd.extract.schema import PROPERTIES, STRATEGIES


@dataclass
class LabelScore:
    label: str
    tp: int = 0
    fp: int = 0
    fn: int = 0

    @property
    def precision(self) -> float:
        return self.tp / (self.tp + self.fp) if self.tp + self.fp else 0.0

    @property
    def recall(self) -> float:
        return self.tp / (self.tp + self.fn) if self.tp + self.fn else 0.0

    @property
    def f1(self) -> float:
        p, r = self.precision, self.recall
        return 2 * p * r / (p + r) if p + r else 0.0


def _score_sets(pred: List[Set[str]], gold: List[Set[str]],
                labels: List[str]) -> Dict[str, LabelScore]:
    scores = {l: LabelScore(l) for l in labels}
    for p, g in zip(pred, gold):
        for l in labels:
            inp, ing = l in p, l in g
            if inp and ing:
                scores[l].tp += 1
            elif inp:
                scores[l].fp += 1
            elif ing:
                scores[l].fn += 1
    return scores


def micro_f1(scores: Dict[str, LabelScore]) -> float:
    tp = sum(s.tp for s in scores.values())
    fp = sum(s.fp for s in scores.values())
    fn = sum(s.fn for s in scores.values())
    p = tp / (tp + fp) if tp + fp else 0.0
    r = tp / (tp + fn) if tp + fn else 0.0
    return 2 * p * r / (p + r) if p + r else 0.0


def evaluate_rules_on_taxonomy(df: pd.DataFrame, limit: int = 0) -> dict:
    """Classify every row's raw text; score vs the reference's labels."""
    if limit:
        df = df.iloc[:limit]
    gold_strat = row_strategies(df)
    gold_props = row_properties(df)
    gold_method = row_method(df).tolist()

    pred_strat: List[Set[str]] = []
    pred_props: List[Set[str]] = []
    pred_method: List[str] = []
    texts = df["Labels"].astype(str).tolist()
    comps = df["Component"].astype(str).tolist()
    repos = df["Repo"].astype(str).tolist()
    from tosem2021_amd.classify.property_lexicon import (
        default_lexicon, property_features)
    from tosem2021_amd.classify.strategy_stack import default_stack
    lex = default_lexicon()
    stack = default_stack()
    for text, comp, repo in zip(texts, comps, repos):
        row = classify_text(text, name="", path=comp)
        feats = None
        if lex is not None or stack is not None:
            feats = property_features(text, comp, repo, row=row)
        rule_set = set(row.strategies())
        if stack is not None:
            pred_strat.append(stack.predict(feats, rule_set))
        else:
            pred_strat.append(rule_set)
        if lex is not None:
            pred_props.append(set(lex.predict(feats)))
        else:
            pred_props.append(set(row.properties()))
        pred_method.append(row.method)

    strat_scores = _score_sets(pred_strat, gold_strat, STRATEGIES)
    prop_scores = _score_sets(pred_props, gold_props, PROPERTIES)
    method_acc = sum(p == g for p, g in zip(pred_method, gold_method)) / len(df)
    res = {
        "n_rows": len(df),
        "strategy": {l: {"precision": s.precision, "recall": s.recall,
                         "f1": s.f1, "support": s.tp + s.fn}
                     for l, s in strat_scores.items()},
        "strategy_micro_f1": micro_f1(strat_scores),
        "property_micro_f1": micro_f1(prop_scores),
        "method_accuracy": method_acc,
        "property_labeler": "lexicon" if lex is not None else "regex",
        "strategy_labeler": "stacked" if stack is not None else "regex",
    }
    if stack is not None:
        odd = [i for i in range(len(df)) if i % 2 == 1]
        strat_ho = _score_sets([pred_strat[i] for i in odd],
                               [gold_strat[i] for i in odd], STRATEGIES)
        res["strategy_micro_f1_heldout"] = micro_f1(strat_ho)
    if lex is not None:
        # the committed lexicon was fit on the even-index gold rows
        # (property_lexicon.py protocol) — report the uncontaminated
        # held-out (odd-row) property score alongside the full-set one
        odd = [i for i in range(len(df)) if i % 2 == 1]
        prop_ho = _score_sets([pred_props[i] for i in odd],
                              [gold_props[i] for i in odd], PROPERTIES)
        res["property_micro_f1_heldout"] = micro_f1(prop_ho)
    return res


def report(result: dict) -> str:
    lines = [f"rows scored: {result['n_rows']}",
             f"strategy micro-F1: {result['strategy_micro_f1']:.3f}",
             f"property micro-F1: {result['property_micro_f1']:.3f}",
             f"method accuracy:   {result['method_accuracy']:.3f}",
             "", f"{'strategy':<28}{'P':>7}{'R':>7}{'F1':>7}{'support':>9}"]
    for l, s in sorted(result["strategy"].items(),
                       key=lambda kv: -kv[1]["support"]):
        lines.append(f"{l:<28}{s['precision']:>7.2f}{s['recall']:>7.2f}"
                     f"{s['f1']:>7.2f}{s['support']:>9}")
    return "\n".join(lines)
