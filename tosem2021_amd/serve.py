"""Model serving endpoint (the corpus' ray `serve/` concern, rebuilt).

A FastAPI app exposing the trained MLTC classifier over HTTP:

    POST /classify   {"texts": ["assertEqual(a, b)", ...],
                      "repos": ["auto_sklearn", ...]   # optional}
        -> per-text strategy/property label lists + stage/method

    GET  /healthz    liveness + model/device info
    GET  /metrics    Prometheus text (utils/metrics.py registry)

Run:  uvicorn tosem2021_amd.serve:build_app --factory --port 8100
(or `python -m tosem2021_amd.serve --ckpt-dir ckpt/` for a dev server).

Batched inference through the same bf16 kernels as training (on GPU) or
the fp32 reference path (CPU); with no checkpoint the app serves the
rule-engine + calibrated-lexicon labels instead, so the endpoint works on
a box with no trained model.  Reference parity: the study's corpus ships
ray `serve/` as its serving stack; this is the
framework's native equivalent for its own flagship model.
"""
from __future__ import annotations

import argparse
import os
from typing import Dict, List, Optional

import torch

from tosem2021_amd.extract.schema import (METHODS, PROPERTIES, STAGES,
                                          STRATEGIES)
from tosem2021_amd.utils.metrics import get_metrics


class ClassifierService:
    """Holds either a trained MLTC (ckpt_dir given) or the rule engine."""

    def __init__(self, ckpt_dir: Optional[str] = None,
                 model: str = "mltc-base", seq: int = 256,
                 threshold: float = 0.5, device: Optional[str] = None,
                 repo_prefix: bool = False):
        self.seq = seq
        self.threshold = threshold
        self.repo_prefix = repo_prefix
        self.trainer = None
        self.tok = None
        self.device = torch.device(device) if device else (
            torch.device("cuda") if torch.cuda.is_available() else
            torch.device("cpu"))
        if ckpt_dir:
            from tosem2021_amd.models.classifier import CONFIGS, MLTCConfig
            from tosem2021_amd.models.tokenizer import CodeTokenizer
            from tosem2021_amd.train import TrainConfig, Trainer
            base = CONFIGS[model]
            cfg = MLTCConfig(**{**base.__dict__, "max_seq": seq})
            tcfg = TrainConfig(model=model, ckpt_dir=ckpt_dir,
                               dtype="bf16" if self.device.type == "cuda"
                               else "f32")
            self.trainer = Trainer(tcfg, device=self.device, model_cfg=cfg)
            if not self.trainer.load_or_init():
                raise FileNotFoundError(f"no checkpoint in {ckpt_dir}")
            self.trainer.model.eval()
            self.tok = CodeTokenizer(cfg.vocab_size)

    @property
    def backend(self) -> str:
        return "mltc" if self.trainer is not None else "rules+lexicon"

    @torch.no_grad()
    def classify(self, texts: List[str],
                 repos: Optional[List[str]] = None) -> List[Dict]:
        repos = repos or [""] * len(texts)
        get_metrics().inc("serve_requests")
        get_metrics().inc("serve_texts", len(texts))
        if self.trainer is None:
            return self._classify_rules(texts, repos)
        inputs = texts
        if self.repo_prefix:
            inputs = [f"REPO_{r} {t}" for r, t in zip(repos, texts)]
        toks, mask = self.tok.encode_batch(inputs, self.seq,
                                           device=self.device)
        logits = self.trainer.model(toks, mask)
        sp = torch.sigmoid(logits["strategy"].float()) > self.threshold
        pp = torch.sigmoid(logits["property"].float()) > self.threshold
        stg = logits["stage"].float().argmax(-1)
        mth = logits["method"].float().argmax(-1)
        out = []
        for i in range(len(texts)):
            out.append({
                "strategies": [STRATEGIES[j] for j in range(len(STRATEGIES))
                               if bool(sp[i, j])],
                "properties": [PROPERTIES[j] for j in range(len(PROPERTIES))
                               if bool(pp[i, j])],
                "stage": STAGES[int(stg[i])],
                "method": METHODS[int(mth[i])],
            })
        return out

    def _classify_rules(self, texts: List[str], repos: List[str]
                        ) -> List[Dict]:
        from tosem2021_amd.classify.property_lexicon import apply_to_row
        from tosem2021_amd.classify.rules import classify_text
        out = []
        for text, repo in zip(texts, repos):
            row = classify_text(text)
            apply_to_row(row, text, "", repo)
            out.append({
                "strategies": row.strategies(),
                "properties": row.properties(),
                "stage": row.category,
                "method": row.method,
            })
        return out


try:  # pydantic model at module scope (PEP-563 string annotations must
    # resolve in the module namespace for FastAPI's signature inspection)
    from pydantic import BaseModel as _BaseModel

    class ClassifyRequest(_BaseModel):
        texts: List[str]
        repos: Optional[List[str]] = None
except ImportError:  # pragma: no cover - fastapi/pydantic absent
    ClassifyRequest = None


def build_app(ckpt_dir: Optional[str] = None, **kw):
    """FastAPI app factory (also used by the tests via TestClient)."""
    from fastapi import FastAPI
    from fastapi.responses import PlainTextResponse

    svc = ClassifierService(ckpt_dir=ckpt_dir or
                            os.environ.get("TOSEM_CKPT_DIR") or None, **kw)
    app = FastAPI(title="tosem2021_amd classifier")

    @app.post("/classify")
    def classify(req: ClassifyRequest):
        if req.repos is not None and len(req.repos) != len(req.texts):
            return {"error": "repos must match texts length"}
        return {"backend": svc.backend,
                "results": svc.classify(req.texts, req.repos)}

    @app.get("/healthz")
    def healthz():
        return {"ok": True, "backend": svc.backend,
                "device": str(svc.device)}

    @app.get("/metrics", response_class=PlainTextResponse)
    def metrics():
        return get_metrics().prometheus_text()

    return app


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--ckpt-dir", default=None)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8100)
    args = ap.parse_args()
    import uvicorn
    uvicorn.run(build_app(args.ckpt_dir), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
