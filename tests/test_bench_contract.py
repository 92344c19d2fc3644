"""bench.py contract: runs on CPU with a tiny config and prints one valid
JSON line with the driver-required keys."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--model", "mltc-tiny",
         "--batch", "2", "--seq", "32", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.strip().splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["metric"] == "train_tokens_per_s"
    assert d["value"] > 0
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["config"]["parallelism"] == "dp1"


def test_bench_torchrun_world2_cpu():
    """Rehearse the driver's multi-GPU launch on CPU (gloo, ws=2)."""
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29612", os.path.join(REPO, "bench.py"),
         "--model", "mltc-tiny", "--batch", "2", "--seq", "32",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=900, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.strip().splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 4
    assert d["n_gpus"] == 2 or d["n_gpus"] == 1  # CPU fallback reports --gpus
