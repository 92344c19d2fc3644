"""Deterministic hashing code tokenizer for the MLTC classifier.

No trained vocabulary (the boxes have no network): identifiers/operators are
split with a code-aware regex and mapped to ids by a stable FNV-1a hash into
the embedding table, with a small reserved id space.  Deterministic across
processes and machines (no PYTHONHASHSEED dependence).
"""
from __future__ import annotations

import re
from typing import List

import torch

PAD, CLS, UNK = 0, 1, 2
N_RESERVED = 8

_TOKEN_RE = re.compile(
    r"[A-Za-z_][A-Za-z0-9_]*"      # identifiers
    r"|\d+\.\d+|\d+"               # numbers
    r"|==|!=|<=|>=|->|\*\*|//|&&|\|\||[\(\)\[\]\{\}<>=+\-*/%.,:;!&|@#'\"]"
)

_CAMEL_RE = re.compile(r"(?<=[a-z0-9])(?=[A-Z])|_")


def _fnv1a(s: str) -> int:
    h = 0xcbf29ce484222325
    for ch in s.encode("utf-8", errors="replace"):
        h ^= ch
        h = (h * 0x100000001b3) & 0xFFFFFFFFFFFFFFFF
    return h


class CodeTokenizer:
    def __init__(self, vocab_size: int = 32768, split_subwords: bool = True):
        assert vocab_size > N_RESERVED
        self.vocab_size = vocab_size
        self.split_subwords = split_subwords

    def tokens(self, text: str) -> List[str]:
        out: List[str] = []
        for tok in _TOKEN_RE.findall(text or ""):
            if self.split_subwords and (tok[0].isalpha() or tok[0] == "_"):
                parts = [p for p in _CAMEL_RE.split(tok) if p]
                out.extend(p.lower() for p in parts) if len(parts) > 1 \
                    else out.append(tok.lower())
            else:
                out.append(tok)
        return out

    def token_id(self, tok: str) -> int:
        return N_RESERVED + _fnv1a(tok) % (self.vocab_size - N_RESERVED)

    def encode(self, text: str, max_len: int) -> List[int]:
        ids = [CLS] + [self.token_id(t) for t in self.tokens(text)]
        return ids[:max_len]

    def encode_batch(self, texts: List[str], max_len: int,
                     device="cpu") -> tuple:
        """Returns (tokens [B, L] int64, mask [B, L] bool)."""
        rows = [self.encode(t, max_len) for t in texts]
        L = max(max(len(r) for r in rows), 1) if rows else 1
        # pad to 64 (flash kernel granularity) when it fits max_len, else 8
        L64 = (L + 63) // 64 * 64
        L = L64 if L64 <= max_len else (L + 7) // 8 * 8
        toks = torch.full((len(rows), L), PAD, dtype=torch.long)
        mask = torch.zeros(len(rows), L, dtype=torch.bool)
        for i, r in enumerate(rows):
            toks[i, :len(r)] = torch.tensor(r, dtype=torch.long)
            mask[i, :len(r)] = True
        return toks.to(device), mask.to(device)
