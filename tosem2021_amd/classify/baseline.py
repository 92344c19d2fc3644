"""Linear baseline classifier (bag-of-hashed-tokens + logistic regression).

The middle rung of the classifier ladder the framework ships:
rule engine (classify/rules.py) < linear baseline (this) < MLTC transformer
(classify/neural.py).  CPU-only, seconds to train; uses the same hashing
tokenizer as MLTC so the comparison isolates the model class.
"""
from __future__ import annotations

from typing import Dict, List

import numpy as np

from tosem2021_amd.classify.agreement import LabelScore, micro_f1
from tosem2021_amd.data.dataset import TaxonomyDataset
from tosem2021_amd.extract.schema import METHODS, PROPERTIES, STRATEGIES
from tosem2021_amd.models.tokenizer import CodeTokenizer


def _features(texts: List[str], tok: CodeTokenizer, dim: int) -> np.ndarray:
    X = np.zeros((len(texts), dim), dtype=np.float32)
    for i, t in enumerate(texts):
        for tk in tok.tokens(t):
            X[i, tok.token_id(tk) % dim] += 1.0
    norm = np.linalg.norm(X, axis=1, keepdims=True)
    return X / np.maximum(norm, 1e-6)


def train_linear_baseline(taxonomy_path: str, dim: int = 4096,
                          val_frac: float = 0.1, seed: int = 0,
                          max_iter: int = 200) -> dict:
    from sklearn.linear_model import LogisticRegression

    from tosem2021_amd.analyze.taxonomy import load_taxonomy

    df = load_taxonomy(taxonomy_path)
    full = TaxonomyDataset.from_taxonomy(df)
    train_ds, val_ds = full.split(val_frac=val_frac, seed=seed)
    tok = CodeTokenizer(32768)
    Xtr = _features(train_ds.texts, tok, dim)
    Xva = _features(val_ds.texts, tok, dim)

    def fit_multilabel(y_tr, y_va, labels) -> Dict[str, LabelScore]:
        scores = {}
        for j, name in enumerate(labels):
            yj = y_tr[:, j]
            s = LabelScore(name)
            if yj.sum() < 4:  # too few positives to fit
                s.fn = int(y_va[:, j].sum())
                scores[name] = s
                continue
            clf = LogisticRegression(max_iter=max_iter, C=4.0,
                                     class_weight="balanced")
            clf.fit(Xtr, yj)
            pred = clf.predict(Xva)
            gold = y_va[:, j]
            s.tp = int(((pred == 1) & (gold == 1)).sum())
            s.fp = int(((pred == 1) & (gold == 0)).sum())
            s.fn = int(((pred == 0) & (gold == 1)).sum())
            scores[name] = s
        return scores

    strat = fit_multilabel(train_ds.strategy.numpy().astype(int),
                           val_ds.strategy.numpy().astype(int), STRATEGIES)
    props = fit_multilabel(train_ds.property_.numpy().astype(int),
                           val_ds.property_.numpy().astype(int), PROPERTIES)

    y_m = train_ds.method.numpy()
    if len(np.unique(y_m)) < 2:
        pred_m = np.full(len(val_ds), y_m[0] if len(y_m) else 0)
    else:
        from sklearn.linear_model import LogisticRegression as LR
        mclf = LR(max_iter=max_iter, C=4.0)
        mclf.fit(Xtr, y_m)
        pred_m = mclf.predict(Xva)
    macc = float((pred_m == val_ds.method.numpy()).mean())

    return {
        "n_train": len(train_ds),
        "n_val": len(val_ds),
        "strategy_micro_f1": round(micro_f1(strat), 4),
        "property_micro_f1": round(micro_f1(props), 4),
        "method_accuracy": round(macc, 4),
        "feature_dim": dim,
    }
