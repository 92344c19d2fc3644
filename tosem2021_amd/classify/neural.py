"""Learned taxonomy classifier: train/evaluate MLTC on labeled taxonomy rows.

The one place the reference's capability legitimately meets the GPU
(SURVEY.md §7 layer 3): fine-tune the transformer classifier on the study's
9,685 human-labeled rows and beat the rule engine's agreement scores.
"""
from __future__ import annotations

import time
from typing import Dict, Optional

import torch

from tosem2021_amd.analyze.taxonomy import load_taxonomy
from tosem2021_amd.classify.agreement import LabelScore, micro_f1
from tosem2021_amd.data.dataset import TaxonomyDataset
from tosem2021_amd.extract.schema import METHODS, PROPERTIES, STRATEGIES
from tosem2021_amd.models.classifier import CONFIGS, MLTCConfig
from tosem2021_amd.models.tokenizer import CodeTokenizer
from tosem2021_amd.train import TrainConfig, Trainer


@torch.no_grad()
def _head_probs(trainer: Trainer, ds: TaxonomyDataset, tok: CodeTokenizer,
                seq: int, batch: int = 64):
    """Collect sigmoid probabilities and gold multi-hot labels per head."""
    model = trainer.model
    model.eval()
    device = trainer.device
    probs = {"strategy": [], "property": []}
    gold = {"strategy": [], "property": []}
    for toks, mask, labels in ds.batches(tok, batch, seq, device=device,
                                         shuffle=False):
        logits = model(toks, mask)
        for h in probs:
            probs[h].append(torch.sigmoid(logits[h].float()).cpu())
            gold[h].append(labels[h].cpu())
    model.train()
    return ({h: torch.cat(v) for h, v in probs.items()},
            {h: torch.cat(v) for h, v in gold.items()})


def tune_thresholds(trainer: Trainer, ds: TaxonomyDataset, tok: CodeTokenizer,
                    seq: int) -> Dict[str, torch.Tensor]:
    """Per-class decision thresholds maximizing F1 on `ds` (use the TRAIN
    split here; evaluate() then reports on the held-out split)."""
    probs, gold = _head_probs(trainer, ds, tok, seq)
    grid = torch.arange(0.05, 0.95, 0.05)
    out = {}
    for h in probs:
        p, g = probs[h], gold[h] > 0.5
        ths = torch.full((p.shape[1],), 0.5)
        for j in range(p.shape[1]):
            best_f1, best_t = -1.0, 0.5
            for t in grid:
                pred = p[:, j] > t
                tp = int((pred & g[:, j]).sum())
                fp = int((pred & ~g[:, j]).sum())
                fn = int((~pred & g[:, j]).sum())
                f1 = 2 * tp / max(2 * tp + fp + fn, 1)
                if f1 > best_f1:
                    best_f1, best_t = f1, float(t)
            ths[j] = best_t
        out[h] = ths
    return out


@torch.no_grad()
def evaluate(trainer: Trainer, ds: TaxonomyDataset, tok: CodeTokenizer,
             seq: int, batch: int = 64, threshold: float = 0.5,
             thresholds: Optional[Dict[str, torch.Tensor]] = None
             ) -> Dict[str, float]:
    model = trainer.model
    model.eval()
    device = trainer.device
    strat_scores = {l: LabelScore(l) for l in STRATEGIES}
    prop_scores = {l: LabelScore(l) for l in PROPERTIES}
    method_hits = 0
    n = 0
    for toks, mask, labels in ds.batches(tok, batch, seq, device=device,
                                         shuffle=False):
        logits = model(toks, mask)
        th_s = thresholds["strategy"].to(device) if thresholds else threshold
        th_p = thresholds["property"].to(device) if thresholds else threshold
        sp = (torch.sigmoid(logits["strategy"].float()) > th_s)
        pp = (torch.sigmoid(logits["property"].float()) > th_p)
        mp = logits["method"].float().argmax(-1)
        gs = labels["strategy"] > 0.5
        gp = labels["property"] > 0.5
        for i, l in enumerate(STRATEGIES):
            s = strat_scores[l]
            s.tp += int((sp[:, i] & gs[:, i]).sum())
            s.fp += int((sp[:, i] & ~gs[:, i]).sum())
            s.fn += int((~sp[:, i] & gs[:, i]).sum())
        for i, l in enumerate(PROPERTIES):
            s = prop_scores[l]
            s.tp += int((pp[:, i] & gp[:, i]).sum())
            s.fp += int((pp[:, i] & ~gp[:, i]).sum())
            s.fn += int((~pp[:, i] & gp[:, i]).sum())
        method_hits += int((mp == labels["method"]).sum())
        n += toks.shape[0]
    model.train()
    return {
        "strategy_micro_f1": micro_f1(strat_scores),
        "property_micro_f1": micro_f1(prop_scores),
        "method_accuracy": method_hits / max(n, 1),
        "n_eval": n,
    }


def _load_dataset(taxonomy_path: str, cfg: MLTCConfig) -> TaxonomyDataset:
    if taxonomy_path.endswith(".pt"):
        ds = TaxonomyDataset.from_tensor_file(taxonomy_path)
    else:
        df = load_taxonomy(taxonomy_path)
        ds = TaxonomyDataset.from_taxonomy(df)
    if ds._tokens is not None and int(ds._tokens.max()) >= cfg.vocab_size:
        # prepared file was tokenized for a bigger vocab (e.g. mltc-base);
        # fold ids into this config's hash space deterministically
        from tosem2021_amd.models.tokenizer import N_RESERVED
        t = ds._tokens
        big = t >= N_RESERVED
        ds._tokens = torch.where(
            big, (t - N_RESERVED) % (cfg.vocab_size - N_RESERVED) + N_RESERVED,
            t)
    return ds


def _run_steps(trainer: Trainer, ds: TaxonomyDataset, tok: CodeTokenizer,
               batch: int, seq: int, steps: int, seed: int, device,
               eval_every: int = 0, val_ds: Optional[TaxonomyDataset] = None,
               tag: str = "", augment: float = 0.0) -> list:
    losses = []
    epoch = 0
    it = iter(())
    aug_keep = aug_gen = None
    if augment > 0:
        from tosem2021_amd.data.augment import augment_tokens, protected_ids
        aug_keep = protected_ids(tok)
        aug_gen = torch.Generator().manual_seed(seed * 7919 + 13)
    while trainer.step_num < steps:
        try:
            toks, mask, labels = next(it)
        except StopIteration:
            it = ds.batches(tok, batch, seq, device=device, shuffle=True,
                            seed=seed + epoch, drop_last=True)
            epoch += 1
            continue
        if augment > 0:
            from tosem2021_amd.data.augment import augment_tokens
            toks = augment_tokens(toks, augment, aug_keep,
                                  tok.vocab_size, aug_gen)
        losses.append(trainer.step(toks, mask, labels))
        if eval_every and val_ds is not None and \
                trainer.step_num % eval_every == 0:
            ev = evaluate(trainer, val_ds, tok, seq)
            print(f"{tag}step {trainer.step_num} loss {losses[-1]:.4f} "
                  f"strategyF1 {ev['strategy_micro_f1']:.3f} "
                  f"propF1 {ev['property_micro_f1']:.3f} "
                  f"methodAcc {ev['method_accuracy']:.3f}")
    return losses, epoch


def train_classifier(taxonomy_path: str, model: str = "mltc-base",
                     steps: int = 500, batch: int = 32, seq: int = 256,
                     lr: float = 3e-4, ckpt_dir: Optional[str] = None,
                     resume: bool = False, device: Optional[str] = None,
                     eval_every: int = 0, seed: int = 0,
                     dropout: float = 0.0,
                     pretrain_path: Optional[str] = None,
                     pretrain_steps: int = 0,
                     focal_gamma_property: float = 0.0,
                     label_smoothing: float = 0.0,
                     dump_probs_path: Optional[str] = None,
                     augment: float = 0.0,
                     split_seed: Optional[int] = None) -> dict:
    """Fine-tune MLTC on `taxonomy_path`'s labeled rows.

    With `pretrain_path`/`pretrain_steps`, first train on that (typically
    mined, rule-labeled) taxonomy, then warm-start the fine-tune from the
    pretrained weights with a fresh optimizer + LR schedule — the mined
    corpus stands in for the study's unlabeled test population (reference
    RQs/taxonomy_test2.csv is only its hand-labeled sample).
    """
    dev = torch.device(device) if device else (
        torch.device("cuda") if torch.cuda.is_available() else
        torch.device("cpu"))
    base = CONFIGS[model]
    cfg = MLTCConfig(**{**base.__dict__, "max_seq": seq,
                     "dropout": dropout})
    full = _load_dataset(taxonomy_path, cfg)
    # split_seed pins the train/val partition independently of the training
    # seed, so differently-seeded runs share one val set and their dumped
    # probs can be ensembled (docs/ROADMAP.md: shared-split k-fold)
    train_ds, val_ds = full.split(
        val_frac=0.1, seed=seed if split_seed is None else split_seed)
    tok = CodeTokenizer(cfg.vocab_size)
    tcfg = TrainConfig(model=model, lr=lr, warmup_steps=min(50, steps // 10),
                       total_steps=steps, ckpt_dir=ckpt_dir,
                       dtype="bf16" if dev.type == "cuda" else "f32")
    trainer = Trainer(tcfg, device=dev, model_cfg=cfg)
    trainer.model.set_pos_weights(
        {k: v.to(dev) for k, v in train_ds.pos_weights().items()})
    if focal_gamma_property or label_smoothing:
        trainer.model.set_loss_options(
            focal_gamma={"property": focal_gamma_property},
            label_smoothing=label_smoothing)
    if resume and ckpt_dir:
        trainer.load_or_init()

    t0 = time.time()
    pretrain_time = 0.0
    if pretrain_path and pretrain_steps > 0 and trainer.step_num == 0:
        pre_ds = _load_dataset(pretrain_path, cfg)
        pre_cfg = TrainConfig(model=model, lr=lr,
                              warmup_steps=min(50, pretrain_steps // 10),
                              total_steps=pretrain_steps,
                              dtype=tcfg.dtype)
        pre_tr = Trainer(pre_cfg, device=dev, model_cfg=cfg)
        pre_tr.model.set_pos_weights(
            {k: v.to(dev) for k, v in pre_ds.pos_weights().items()})
        _run_steps(pre_tr, pre_ds, tok, batch, seq, pretrain_steps, seed,
                   dev, eval_every, val_ds, tag="pre ")[0]
        # warm-start: pretrained weights, fresh optimizer state + schedule
        trainer.flat.flat.copy_(pre_tr.flat.flat)
        trainer.flat.master.copy_(pre_tr.flat.master)
        del pre_tr
        pretrain_time = time.time() - t0

    losses, epoch = _run_steps(trainer, train_ds, tok, batch, seq, steps,
                               seed, dev, eval_every, val_ds,
                               augment=augment)
    train_time = time.time() - t0
    ev = evaluate(trainer, val_ds, tok, seq)
    ev_lo = evaluate(trainer, val_ds, tok, seq, threshold=0.3)
    ev["strategy_micro_f1_t0.3"] = ev_lo["strategy_micro_f1"]
    ev["property_micro_f1_t0.3"] = ev_lo["property_micro_f1"]
    ths = tune_thresholds(trainer, train_ds, tok, seq)
    ev_tuned = evaluate(trainer, val_ds, tok, seq, thresholds=ths)
    ev["strategy_micro_f1_tuned"] = round(ev_tuned["strategy_micro_f1"], 4)
    ev["property_micro_f1_tuned"] = round(ev_tuned["property_micro_f1"], 4)
    if dump_probs_path:
        # sigmoid probs + gold for the val AND train splits — enables
        # offline ensembling/blending without the GPU
        vp, vg = _head_probs(trainer, val_ds, tok, seq)
        tp_, tg = _head_probs(trainer, train_ds, tok, seq)
        torch.save({"val_probs": vp, "val_gold": vg,
                    "train_probs": tp_, "train_gold": tg,
                    "seed": seed}, dump_probs_path)
    if ckpt_dir:
        trainer.save()
    return {
        "steps": trainer.step_num,
        "epochs": epoch,
        "train_time_s": round(train_time, 2),
        "pretrain_time_s": round(pretrain_time, 2),
        "pretrain_steps": pretrain_steps if pretrain_path else 0,
        "final_loss": losses[-1] if losses else None,
        "loss_first10_mean": (sum(losses[:10]) / min(len(losses), 10)
                              if losses else None),
        **{k: (round(v, 4) if isinstance(v, float) else v)
           for k, v in ev.items()},
    }


@torch.no_grad()
def apply_classifier(ckpt_dir: str, taxonomy_path: str, out_csv: str,
                     model: str = "mltc-base", seq: int = 256,
                     batch: int = 64, threshold: float = 0.5,
                     device: Optional[str] = None,
                     repo_prefix: bool = False) -> str:
    """Label the rows of a (mined) taxonomy CSV with a trained MLTC model and
    rewrite the 41-column CSV with the predicted labels."""
    from tosem2021_amd.extract.schema import (
        METHODS, PROPERTIES, STAGES, STRATEGIES, STRATEGY_TO_COLUMNS,
        TestCaseRow)
    from tosem2021_amd.pipeline import write_taxonomy_csv

    dev = torch.device(device) if device else (
        torch.device("cuda") if torch.cuda.is_available() else
        torch.device("cpu"))
    df = load_taxonomy(taxonomy_path)
    base = CONFIGS[model]
    cfg = MLTCConfig(**{**base.__dict__, "max_seq": seq})
    tok = CodeTokenizer(cfg.vocab_size)
    tcfg = TrainConfig(model=model, ckpt_dir=ckpt_dir,
                       dtype="bf16" if dev.type == "cuda" else "f32")
    trainer = Trainer(tcfg, device=dev, model_cfg=cfg)
    assert trainer.load_or_init(), f"no checkpoint found in {ckpt_dir}"
    trainer.model.eval()

    texts = (df["Labels"].astype(str) + " | " +
             df["Component"].astype(str)).tolist()
    if repo_prefix:
        # models trained on repo-prefixed text (artifacts/
        # mltc_train_repo_s256.pt) expect the same context token
        texts = [f"REPO_{r} {t}"
                 for r, t in zip(df["Repo"].astype(str), texts)]
    rows = []
    for lo in range(0, len(texts), batch):
        chunk = texts[lo:lo + batch]
        toks, mask = tok.encode_batch(chunk, seq, device=dev)
        logits = trainer.model(toks, mask)
        sp = torch.sigmoid(logits["strategy"].float()) > threshold
        pp = torch.sigmoid(logits["property"].float()) > threshold
        stg = logits["stage"].float().argmax(-1)
        mth = logits["method"].float().argmax(-1)
        for i in range(len(chunk)):
            src = df.iloc[lo + i]
            row = TestCaseRow(
                index=lo + i + 1, labels=str(src["Labels"]),
                component=str(src["Component"]), repo=str(src["Repo"]),
                category=STAGES[int(stg[i])], category2=STAGES[int(stg[i])])
            flags = {}
            for j, name in enumerate(STRATEGIES):
                if not bool(sp[i, j]):
                    continue
                spec = STRATEGY_TO_COLUMNS[name]
                flags.update(spec.get("flags", {}))
                if "error_type" in spec:
                    row.error_type = spec["error_type"]
                if "approximation_type" in spec:
                    row.approximation_type = spec["approximation_type"]
                if "checks_type" in spec:
                    row.checks_type = spec["checks_type"]
            m = METHODS[int(mth[i])]
            if m == "regression":
                flags["regression"] = 1
            elif m == "integration":
                flags["Integration"] = 1
            elif m == "end_to_end":
                flags["end_to_end"] = 1
            if not flags:
                flags["None_above"] = 1
            row.flags = flags
            props = [PROPERTIES[j] for j in range(len(PROPERTIES))
                     if bool(pp[i, j])]
            # row_properties reads all four of Model/Data/Code/Oracle — spill
            # predicted properties across them so nothing is dropped.
            for slot, val in zip(("model", "data", "code", "oracle"), props):
                setattr(row, slot, val)
            rows.append(row)
    trainer.model.train()
    return write_taxonomy_csv(rows, out_csv)
