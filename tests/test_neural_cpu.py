

def test_augment_tokens_invariants():
    """Assertion-text augmentation: reserved + protected ids unchanged,
    per-example consistency, in-range outputs, p=0 identity."""
    import torch
    from tosem2021_amd.data.augment import augment_tokens, protected_ids
    from tosem2021_amd.models.tokenizer import N_RESERVED, CodeTokenizer
    tok = CodeTokenizer(4096)
    keep = protected_ids(tok)
    g = torch.Generator().manual_seed(0)
    toks = torch.randint(0, 4096, (6, 48), generator=g)
    out = augment_tokens(toks, 0.4, keep, 4096, g)
    res = toks < N_RESERVED
    assert torch.equal(out[res], toks[res])
    pro = torch.isin(toks, keep)
    assert torch.equal(out[pro], toks[pro])
    assert int(out.min()) >= 0 and int(out.max()) < 4096
    for i in range(toks.shape[0]):
        seen = {}
        for a, b in zip(toks[i].tolist(), out[i].tolist()):
            assert seen.setdefault(a, b) == b
    assert (out != toks).any()
    assert augment_tokens(toks, 0.0, keep, 4096, g) is toks


def test_split_seed_pins_partition():
    """split_seed gives differently-seeded runs the same val rows."""
    import torch
    from tosem2021_amd.data.dataset import TaxonomyDataset
    ds = TaxonomyDataset(
        [f"assert x == {i}" for i in range(50)],
        torch.zeros(50, 19), torch.zeros(50, 21),
        torch.zeros(50, dtype=torch.long), torch.zeros(50, dtype=torch.long))
    _, va1 = ds.split(val_frac=0.2, seed=7)
    _, va2 = ds.split(val_frac=0.2, seed=7)
    assert va1.texts == va2.texts
