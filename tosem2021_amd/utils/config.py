"""YAML experiment configuration (capability parity with the corpus' config
systems the study measured — SURVEY.md §5: absl flags / gflags+proto conf /
nni YAML experiment config): load YAML into the framework's dataclass
configs with validation and dotted-key overrides."""
from __future__ import annotations

import dataclasses
from typing import Any, Dict, Optional

import yaml

from tosem2021_amd.models.classifier import CONFIGS, MLTCConfig
from tosem2021_amd.train import TrainConfig


@dataclasses.dataclass
class ExperimentConfig:
    name: str = "experiment"
    corpus_root: Optional[str] = None
    projects: Optional[list] = None
    taxonomy_csv: Optional[str] = None
    out_dir: str = "out"
    train: TrainConfig = dataclasses.field(default_factory=TrainConfig)
    model_cfg: Optional[MLTCConfig] = None


def _apply(dc, data: Dict[str, Any], path: str):
    valid = {f.name: f for f in dataclasses.fields(dc)}
    for k, v in data.items():
        if k not in valid:
            raise KeyError(f"unknown config key: {path}{k}")
        cur = getattr(dc, k)
        if dataclasses.is_dataclass(cur) and isinstance(v, dict):
            _apply(cur, v, f"{path}{k}.")
        else:
            setattr(dc, k, v)


def load_config(path: str) -> ExperimentConfig:
    with open(path) as f:
        data = yaml.safe_load(f) or {}
    cfg = ExperimentConfig()
    model_name = data.get("train", {}).get("model")
    if isinstance(data.get("model_cfg"), dict):
        base = CONFIGS.get(model_name or "mltc-base", CONFIGS["mltc-base"])
        cfg.model_cfg = MLTCConfig(**{**base.__dict__, **data.pop("model_cfg")})
    _apply(cfg, data, "")
    return cfg


def apply_overrides(cfg: ExperimentConfig, overrides: Dict[str, Any]
                    ) -> ExperimentConfig:
    """Dotted-key overrides, e.g. {"train.lr": 1e-4}."""
    for key, v in overrides.items():
        obj = cfg
        parts = key.split(".")
        for p in parts[:-1]:
            obj = getattr(obj, p)
        if not hasattr(obj, parts[-1]):
            raise KeyError(f"unknown override key: {key}")
        setattr(obj, parts[-1], v)
    return cfg
