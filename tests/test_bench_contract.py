"""The driver depends on bench.py's CLI + JSON output contract; verify it
end-to-end on the torch backend (CPU, tiny grid)."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [
            sys.executable,
            os.path.join(REPO, "bench.py"),
            "--backend",
            "torch",
            "--gpus",
            "1",
            "--per-gpu",
            "32",
            "--steps",
            "2",
            "--warmup",
            "1",
        ],
        capture_output=True,
        text=True,
        timeout=300,
        cwd=REPO,
    )
    assert out.returncode == 0, out.stderr
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    for key in (
        "metric",
        "value",
        "unit",
        "n_gpus",
        "steps",
        "warmup",
        "ms_per_step",
        "higher_is_better",
        "scaling",
        "vs_baseline",
        "dtype",
        "data",
        "config",
    ):
        assert key in d, key
    assert d["metric"] == "jacobi3d_cell_updates_per_s"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["config"]["model"] == "jacobi3d"


def test_weak_dims():
    sys.path.insert(0, REPO)
    import importlib

    bench = importlib.import_module("bench")
    for n, want in [(1, [1, 1, 1]), (2, [2, 1, 1]), (4, [2, 2, 1]), (8, [2, 2, 2]), (6, [3, 2, 1])]:
        assert bench.weak_dims(n) == want, n
