"""Token-level assertion-text augmentation for MLTC training.

The tokenizer is a stable-hash map (models/tokenizer.py), so "identifier
renaming" — the classic robustness augmentation for code models — becomes a
CONSISTENT remap of a random subset of an example's token ids to other ids
in the same hash space: selection and destination are both functions of the
id (with per-example random coefficients), so the same identifier keeps the
same new id everywhere in the example — co-occurrence structure survives,
the specific surface name does not.  Semantics-bearing tokens (the
assert/expect vocabulary, comparison words, common test-API subwords) are
protected so the label signal survives; everything else (test-specific
identifiers, module names, literals) is fair game.

Mirrors reference capability only in spirit: the reference ships no
training code (SURVEY.md §2 — data-only replication package).
"""
from __future__ import annotations

from typing import Optional, Set

import torch

from tosem2021_amd.models.tokenizer import N_RESERVED, CodeTokenizer

# Subword forms (the tokenizer lower-cases and splits camelCase) whose ids
# must never be renamed: they carry the strategy/property signal.
_PROTECTED_WORDS = """
assert assertequal asserttrue assertfalse assertraises assertin assertis
assertnone assertalmostequal assertgreater assertless assertthat expect
expected to be equal equals eq ne gt lt ge le true false none null not is in
raise raises raised throw throws thrown error exception valueerror typeerror
keyerror runtimeerror indexerror close almost approx near greater less than
within len length shape size dtype type isinstance instance count empty
contains contain match matches called once mock patch value values range
status ok fail failed pass passed check checks validate valid invalid
np numpy torch tf self pytest unittest test tests
allclose array_equal equal_nan decimal places rel abs tol atol rtol
== != < > <= >= | same diff different bound bounds min max zero nan inf
""".split()

_P_MOD = 997   # prime selection modulus (hash-based bernoulli per id)


def protected_ids(tok: CodeTokenizer) -> torch.Tensor:
    ids = sorted({tok.token_id(w) for w in _PROTECTED_WORDS})
    return torch.tensor([i for i in ids if i >= N_RESERVED],
                        dtype=torch.long)


def augment_tokens(toks: torch.Tensor, p_rename: float,
                   keep: torch.Tensor, vocab_size: int,
                   generator: Optional[torch.Generator] = None
                   ) -> torch.Tensor:
    """Consistently rename ~`p_rename` of each example's non-reserved,
    non-protected token ids.  Fully vectorized: selection is a hashed
    bernoulli of the id and the remap an affine map of the id, both with
    per-example random coefficients, so each id is handled identically at
    every occurrence within an example.  `toks` is [B, L] int64 (any
    device; coefficients are drawn on CPU for determinism)."""
    if p_rename <= 0:
        return toks
    B = toks.shape[0]
    lo, span = N_RESERVED, vocab_size - N_RESERVED
    coef = torch.randint(0, 2 ** 30, (B, 4), generator=generator)
    coef = coef.to(toks.device)
    c, d, a, b = (coef[:, 0:1] | 1), coef[:, 1:2], (coef[:, 2:3] | 1), \
        coef[:, 3:4]
    sel = ((toks * c + d) % _P_MOD) < int(p_rename * _P_MOD)
    sel &= toks >= lo
    sel &= ~torch.isin(toks, keep.to(toks.device))
    new = lo + ((toks - lo) * a + b) % span
    return torch.where(sel, new, toks)
