"""The driver depends on bench.py's output contract (BASELINE.json): exactly one
JSON line on stdout with the required keys.  This lane catches contract breaks
before the driver does."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {
    "metric": str, "value": float, "unit": str, "n_gpus": int, "steps": int,
    "warmup": int, "ms_per_step": float, "higher_is_better": bool, "scaling": str,
    "dtype": str, "data": str, "config": dict,
}


def test_bench_stdout_is_one_contract_json_line():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "1",
         "--network", "FC", "--dataset", "MNIST", "--batch-size", "8",
         "--device", "cpu"],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.strip()]
    assert len(lines) == 1, f"stdout must be EXACTLY one line, got {len(lines)}"
    r = json.loads(lines[0])
    for k, ty in REQUIRED.items():
        assert k in r, f"missing contract key {k}"
        assert isinstance(r[k], ty), (k, type(r[k]))
    assert r["n_gpus"] == 1 and r["steps"] == 3 and r["warmup"] == 1
    assert r["scaling"] == "weak"
    assert r["higher_is_better"] is True
    assert "vs_baseline" in r  # may be float or null for approach=baseline
    cfg = r["config"]
    for k in ("model", "global_batch", "parallelism", "topology"):
        assert k in cfg, f"missing config key {k}"
    # honesty telemetry
    for k in ("final_loss", "skipped_updates"):
        assert k in r, f"missing telemetry key {k}"
