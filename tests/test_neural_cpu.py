

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


def test_ensemble_probs_script(tmp_path):
    """scripts/ensemble_probs.py: shared-gold verification + averaging."""
    import subprocess
    import sys

    import torch
    g = torch.Generator().manual_seed(0)
    gold = {"strategy": (torch.rand(20, 19, generator=g) < 0.2).float(),
            "property": (torch.rand(20, 21, generator=g) < 0.2).float()}
    paths = []
    for s in range(2):
        probs = {h: torch.rand(*gold[h].shape, generator=g) * 0.5 +
                 gold[h] * 0.4 for h in gold}
        p = tmp_path / f"p{s}.pt"
        torch.save({"val_probs": probs, "val_gold": gold, "seed": s}, p)
        paths.append(str(p))
    out = subprocess.run(
        [sys.executable, "scripts/ensemble_probs.py"] + paths,
        capture_output=True, text=True, cwd="/root/repo")
    assert out.returncode == 0, out.stderr
    assert "ENSEMBLE x2" in out.stdout
    # mismatched gold must abort
    bad = {"val_probs": {h: torch.rand(*gold[h].shape) for h in gold},
           "val_gold": {h: 1 - gold[h] for h in gold}, "seed": 9}
    pb = tmp_path / "bad.pt"
    torch.save(bad, pb)
    out2 = subprocess.run(
        [sys.executable, "scripts/ensemble_probs.py", paths[0], str(pb)],
        capture_output=True, text=True, cwd="/root/repo")
    assert out2.returncode != 0
