"""Command-line interface: survey-style pipeline commands.

    python -m tosem2021_amd.cli mine --projects auto-sklearn --out tax.csv
    python -m tosem2021_amd.cli analyze --taxonomy tax.csv --out out/RQs
    python -m tosem2021_amd.cli agreement --taxonomy /root/reference/RQs/taxonomy_test2.csv
    python -m tosem2021_amd.cli golden --ours out/RQs --reference /root/reference/RQs
    python -m tosem2021_amd.cli train --taxonomy ... --steps 200
    python -m tosem2021_amd.cli report --taxonomy tax.csv
"""
from __future__ import annotations

import argparse
import json
import os
import sys


def cmd_mine(args):
    from tosem2021_amd.corpus.registry import PROJECTS, available_projects
    from tosem2021_amd.pipeline import mine
    projects = args.projects or available_projects(args.corpus_root)
    bad = [p for p in projects if p not in PROJECTS]
    if bad:
        sys.exit(f"unknown projects: {bad}; known: {list(PROJECTS)}")
    path = mine(projects, args.out, corpus_root=args.corpus_root,
                languages=tuple(args.languages.split(",")),
                workers=args.workers)
    print(f"wrote {path}")


def cmd_analyze(args):
    from tosem2021_amd.analyze.taxonomy import load_taxonomy
    df = load_taxonomy(args.taxonomy)
    if args.legacy_tables:
        from tosem2021_amd.analyze.tables import write_all
        paths = write_all(df, args.out)
        if not args.no_figures:
            from tosem2021_amd.analyze.figures import write_figures
            paths.update(write_figures(df, args.out))
    else:
        from tosem2021_amd.analyze.mirror import write_mirror
        paths = write_mirror(df, args.out)
    for k, p in paths.items():
        print(f"{k}: {p}")


def cmd_agreement(args):
    from tosem2021_amd.analyze.taxonomy import load_taxonomy
    from tosem2021_amd.classify.agreement import (
        evaluate_rules_on_taxonomy, report)
    df = load_taxonomy(args.taxonomy)
    res = evaluate_rules_on_taxonomy(df, limit=args.limit)
    print(report(res))
    if args.json:
        with open(args.json, "w") as f:
            json.dump(res, f, indent=2)


def cmd_golden(args):
    if args.mirror:
        from tosem2021_amd.analyze.golden_mirror import mirror_diff
        res = mirror_diff(args.ours, args.reference)
    else:
        from tosem2021_amd.analyze.golden import golden_diff
        res = golden_diff(args.ours, args.reference)
    print(json.dumps(res, indent=2))
    sys.exit(0 if res["ok"] else 1)


def cmd_labels(args):
    """Ingest the L2 manual-labeling artifacts (XLSX/CSV sheets + codebooks)
    and report lineage/coverage against the master taxonomy."""
    from tosem2021_amd.analyze.taxonomy import load_taxonomy
    from tosem2021_amd.corpus.labels import (
        codebook_strategy_coverage, lineage_check, load_all_release_sheets,
        load_case_labels, load_codebook)
    sheets = load_all_release_sheets(args.reference)
    cb = load_codebook(args.reference)
    res = {
        "release_sheets": {
            k: {"rows": len(v),
                "with_file_id": sum(1 for r in v if r.file_id is not None),
                "with_test_type": sum(1 for r in v if r.test_type)}
            for k, v in sheets.items()},
        "codebook": codebook_strategy_coverage(cb),
        "case_labels": len(load_case_labels(args.reference)),
    }
    if args.taxonomy:
        df = load_taxonomy(args.taxonomy)
        res["lineage"] = lineage_check(df, sheets)
    print(json.dumps(res, indent=2))
    if args.json:
        with open(args.json, "w") as f:
            json.dump(res, f, indent=2)


def cmd_report(args):
    from tosem2021_amd.analyze.report import summary_report
    from tosem2021_amd.analyze.taxonomy import load_taxonomy
    df = load_taxonomy(args.taxonomy)
    print(summary_report(df))


def cmd_funnel(args):
    from tosem2021_amd.corpus.selection import (
        DEFAULT_ROUNDS, FunnelCriteria, load_metrics, run_funnel)
    df = load_metrics(args.metrics)
    rounds = DEFAULT_ROUNDS
    if args.min_stars or args.min_commits or args.min_contributors:
        rounds = [FunnelCriteria(name="custom", min_stars=args.min_stars,
                                 min_commits=args.min_commits,
                                 min_contributors=args.min_contributors,
                                 min_releases=args.min_releases)]
    outs = run_funnel(df, rounds)
    print(f"input: {len(df)} candidates")
    for crit, out in zip(rounds, outs):
        print(f"round '{crit.name}': {len(out)} survive")
    if args.out:
        os.makedirs(args.out, exist_ok=True)
        for i, out in enumerate(outs):
            p = os.path.join(args.out, f"round_{i + 1}.csv")
            out.to_csv(p, index=False)
            print(f"wrote {p}")


def cmd_classify(args):
    from tosem2021_amd.classify.neural import apply_classifier
    out = apply_classifier(args.ckpt_dir, args.taxonomy, args.out,
                           model=args.model, seq=args.seq,
                           threshold=args.threshold,
                           repo_prefix=args.repo_prefix)
    print(f"wrote {out}")


def cmd_train(args):
    from tosem2021_amd.classify.neural import train_classifier
    res = train_classifier(
        taxonomy_path=args.taxonomy, model=args.model, steps=args.steps,
        batch=args.batch, seq=args.seq, lr=args.lr, ckpt_dir=args.ckpt_dir,
        resume=args.resume, dropout=args.dropout,
        pretrain_path=args.pretrain, pretrain_steps=args.pretrain_steps,
        focal_gamma_property=args.focal_gamma_property,
        label_smoothing=args.label_smoothing, eval_every=args.eval_every,
        seed=args.seed, dump_probs_path=args.dump_probs,
        augment=args.augment, split_seed=args.split_seed)
    print(json.dumps(res, indent=2))


def main(argv=None):
    ap = argparse.ArgumentParser(prog="tosem2021_amd")
    sub = ap.add_subparsers(dest="cmd", required=True)

    p = sub.add_parser("mine", help="mine a corpus into a taxonomy CSV")
    p.add_argument("--projects", nargs="*", default=None)
    p.add_argument("--corpus-root", default=None)
    p.add_argument("--languages", default="python,cpp,ts")
    p.add_argument("--workers", type=int, default=0)
    p.add_argument("--out", required=True)
    p.set_defaults(fn=cmd_mine)

    p = sub.add_parser("analyze", help="regenerate the full RQs/ artifact "
                       "mirror (every shipped CSV + SVG plots)")
    p.add_argument("--taxonomy", required=True)
    p.add_argument("--out", required=True)
    p.add_argument("--no-figures", action="store_true")
    p.add_argument("--legacy-tables", action="store_true",
                   help="emit the round-1 analysis tables instead of the "
                        "file-for-file mirror")
    p.set_defaults(fn=cmd_analyze)

    p = sub.add_parser("agreement", help="score rules vs the reference labels")
    p.add_argument("--taxonomy", required=True)
    p.add_argument("--limit", type=int, default=0)
    p.add_argument("--json", default=None)
    p.set_defaults(fn=cmd_agreement)

    p = sub.add_parser("golden", help="diff regenerated tables vs reference")
    p.add_argument("--ours", required=True)
    p.add_argument("--reference", required=True)
    p.add_argument("--mirror", action="store_true",
                   help="compare the complete file-for-file RQs/ mirror")
    p.set_defaults(fn=cmd_golden)

    p = sub.add_parser("labels", help="ingest the L2 labeling sheets + "
                       "codebooks; report L2->L3 lineage")
    p.add_argument("--reference", default="/root/reference")
    p.add_argument("--taxonomy", default=None,
                   help="master taxonomy CSV for the lineage check")
    p.add_argument("--json", default=None)
    p.set_defaults(fn=cmd_labels)

    p = sub.add_parser("report", help="print a taxonomy summary report")
    p.add_argument("--taxonomy", required=True)
    p.set_defaults(fn=cmd_report)

    p = sub.add_parser("funnel", help="run the repo-selection funnel")
    p.add_argument("--metrics", required=True,
                   help="GitHub-metadata CSV (Repos_metrics_v3.csv schema)")
    p.add_argument("--out", default=None)
    p.add_argument("--min-stars", type=int, default=0)
    p.add_argument("--min-commits", type=int, default=0)
    p.add_argument("--min-contributors", type=int, default=0)
    p.add_argument("--min-releases", type=int, default=0)
    p.set_defaults(fn=cmd_funnel)

    p = sub.add_parser("classify", help="label a taxonomy with a trained model")
    p.add_argument("--taxonomy", required=True)
    p.add_argument("--ckpt-dir", required=True)
    p.add_argument("--out", required=True)
    p.add_argument("--model", default="mltc-base")
    p.add_argument("--seq", type=int, default=256)
    p.add_argument("--threshold", type=float, default=0.5)
    p.add_argument("--repo-prefix", action="store_true",
                   help="model was trained on repo-prefixed text")
    p.set_defaults(fn=cmd_classify)

    p = sub.add_parser("train", help="train the MLTC classifier on a taxonomy")
    p.add_argument("--taxonomy", required=True)
    p.add_argument("--model", default="mltc-base")
    p.add_argument("--steps", type=int, default=500)
    p.add_argument("--batch", type=int, default=32)
    p.add_argument("--seq", type=int, default=256)
    p.add_argument("--lr", type=float, default=3e-4)
    p.add_argument("--ckpt-dir", default=None)
    p.add_argument("--resume", action="store_true")
    p.add_argument("--dropout", type=float, default=0.0)
    p.add_argument("--pretrain", default=None,
                   help="mined taxonomy to pretrain on before fine-tuning")
    p.add_argument("--pretrain-steps", type=int, default=0)
    p.add_argument("--augment", type=float, default=0.0,
                   help="fraction of non-protected token ids consistently "
                        "renamed per example (assertion-text augmentation)")
    p.add_argument("--split-seed", type=int, default=None,
                   help="pin the train/val split independently of --seed "
                        "(shared-split ensembling)")
    p.add_argument("--focal-gamma-property", type=float, default=0.0)
    p.add_argument("--label-smoothing", type=float, default=0.0)
    p.add_argument("--eval-every", type=int, default=0)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--dump-probs", default=None,
                   help="save val/train sigmoid probs + gold for offline "
                        "ensembling")
    p.set_defaults(fn=cmd_train)

    args = ap.parse_args(argv)
    args.fn(args)


if __name__ == "__main__":
    main()
