"""Serving throughput: MLTC classification inference (no_grad, bf16).

Exercises the inference-only paths (hipBLASLt GELU_BIAS epilogue FFN,
flash attention forward) at the apply_classifier serving shape.
Prints one JSON line; auxiliary to the driver's training bench.py contract.
"""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import argparse
import json
import time

import torch

from tosem2021_amd.models.classifier import build_model


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="mltc-base")
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--seq", type=int, default=256)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--graph", action="store_true",
                    help="replay the forward as one captured hipGraph")
    args = ap.parse_args()

    torch.manual_seed(0)
    model = build_model(args.model, dtype=torch.bfloat16).cuda().eval()
    vocab = model.cfg.vocab_size
    toks = torch.randint(4, vocab, (args.batch, args.seq), device="cuda")
    mask = torch.ones(args.batch, args.seq, device="cuda",
                      dtype=torch.bool)

    if args.graph:
        from tosem2021_amd.utils.hipgraph import CapturedForward
        fwd = CapturedForward(model, args.batch, args.seq)
        run = lambda: fwd(toks, mask)  # noqa: E731
    else:
        run = lambda: model(toks, mask)  # noqa: E731
    with torch.no_grad():
        for _ in range(args.warmup):
            logits = run()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            logits = run()
            probs = {h: torch.sigmoid(v.float()) for h, v in logits.items()}
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters

    print(json.dumps({
        "metric": "infer_examples_per_s",
        "value": args.batch / dt,
        "unit": "examples/s",
        "ms_per_batch": dt * 1e3,
        "dtype": "bf16",
        "data": "synthetic",
        "config": {"model": args.model, "batch": args.batch,
                   "seq_len": args.seq, "hipgraph": bool(args.graph)},
    }))


if __name__ == "__main__":
    main()
