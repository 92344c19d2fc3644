"""Taxonomy dataset: the study's 9,685 labeled rows as classifier training
data (tokens + multi-hot strategy/property labels + stage/method classes)."""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Iterator, List, Tuple

import pandas as pd
import torch

from tosem2021_amd.analyze.taxonomy import (
    row_method, row_properties, row_stage, row_strategies)
from tosem2021_amd.extract.schema import (
    METHODS, PROPERTIES, STAGES, STRATEGIES)
from tosem2021_amd.models.tokenizer import CodeTokenizer


@dataclass
class TaxonomyDataset:
    texts: List[str]
    strategy: torch.Tensor   # [N, 19] multi-hot
    property_: torch.Tensor  # [N, 21] multi-hot
    stage: torch.Tensor      # [N] class id
    method: torch.Tensor     # [N] class id

    _tokens = None   # pre-tokenized [N, L] (tensor-file datasets)
    _mask = None

    def __len__(self) -> int:
        return len(self.texts) if self.texts else (
            0 if self._tokens is None else self._tokens.shape[0])

    @classmethod
    def from_tensor_file(cls, path: str) -> "TaxonomyDataset":
        """Load a prepared (tokenized) dataset — see prepare_tensor_file.

        The file stores token ids + label tensors only (derived data, no raw
        text), so GPU boxes can train on the study's labeled rows without the
        reference mount."""
        import torch as _t
        d = _t.load(path, weights_only=True)
        ds = cls([], d["strategy"].float(), d["property"].float(),
                 d["stage"].long(), d["method"].long())
        ds._tokens = d["tokens"].long()
        ds._mask = d["mask"].bool()
        return ds

    def save_tensor_file(self, path: str, tokenizer, max_len: int) -> str:
        import torch as _t
        toks, mask = tokenizer.encode_batch(self.texts, max_len)
        assert int(toks.max()) <= 32767, "int16 token storage needs vocab<=32768"
        _t.save({"tokens": toks.to(_t.int16), "mask": mask,
                 "strategy": self.strategy.to(_t.int8),
                 "property": self.property_.to(_t.int8),
                 "stage": self.stage.to(_t.int16),
                 "method": self.method.to(_t.int8)}, path)
        return path

    def pos_weights(self) -> dict:
        """Inverse-frequency positive weights for the multi-label heads."""
        import torch as _t
        out = {}
        for name, lab in (("strategy", self.strategy),
                          ("property", self.property_)):
            n = lab.shape[0]
            pos = lab.sum(0).clamp(min=1.0)
            out[name] = ((n - pos) / pos).clamp(max=50.0)
        return out

    @classmethod
    def from_taxonomy(cls, df: pd.DataFrame) -> "TaxonomyDataset":
        strat_sets = row_strategies(df)
        prop_sets = row_properties(df)
        stages = row_stage(df).tolist()
        methods = row_method(df).tolist()
        n = len(df)
        strategy = torch.zeros(n, len(STRATEGIES))
        property_ = torch.zeros(n, len(PROPERTIES))
        stage = torch.zeros(n, dtype=torch.long)
        method = torch.zeros(n, dtype=torch.long)
        for i in range(n):
            for s in strat_sets[i]:
                strategy[i, STRATEGIES.index(s)] = 1
            for p in prop_sets[i]:
                property_[i, PROPERTIES.index(p)] = 1
            stage[i] = STAGES.index(stages[i]) if stages[i] in STAGES else \
                STAGES.index("config_utility")
            method[i] = METHODS.index(methods[i])
        texts = (df["Labels"].astype(str) + " | " +
                 df["Component"].astype(str)).tolist()
        return cls(texts, strategy, property_, stage, method)

    def split(self, val_frac: float = 0.1, seed: int = 0
              ) -> Tuple["TaxonomyDataset", "TaxonomyDataset"]:
        g = torch.Generator().manual_seed(seed)
        perm = torch.randperm(len(self), generator=g)
        n_val = int(len(self) * val_frac)
        va, tr = perm[:n_val], perm[n_val:]

        def take(ix):
            ds = TaxonomyDataset(
                [self.texts[i] for i in ix.tolist()] if self.texts else [],
                self.strategy[ix], self.property_[ix],
                self.stage[ix], self.method[ix])
            if self._tokens is not None:
                ds._tokens = self._tokens[ix]
                ds._mask = self._mask[ix]
            return ds
        return take(tr), take(va)

    def batches(self, tokenizer: CodeTokenizer, batch_size: int, max_len: int,
                device="cpu", shuffle: bool = True, seed: int = 0,
                drop_last: bool = False
                ) -> Iterator[Tuple[torch.Tensor, torch.Tensor,
                                    Dict[str, torch.Tensor]]]:
        n = len(self)
        g = torch.Generator().manual_seed(seed)
        order = torch.randperm(n, generator=g) if shuffle else torch.arange(n)
        for lo in range(0, n, batch_size):
            ix = order[lo:lo + batch_size]
            if drop_last and len(ix) < batch_size:
                break
            if self._tokens is not None:
                toks = self._tokens[ix, :max_len].to(device)
                mask = self._mask[ix, :max_len].to(device)
            else:
                toks, mask = tokenizer.encode_batch(
                    [self.texts[i] for i in ix.tolist()], max_len,
                    device=device)
            labels = {
                "strategy": self.strategy[ix].to(device),
                "property": self.property_[ix].to(device),
                "stage": self.stage[ix].to(device),
                "method": self.method[ix].to(device),
            }
            yield toks, mask, labels
