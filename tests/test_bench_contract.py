"""bench.py driver-contract test: flags accepted, one JSON line on stdout
with the required fields/types (run in CPU plumbing mode)."""
import json
import subprocess
import sys

REQUIRED = {
    "metric": str, "value": float, "unit": str, "n_gpus": int, "steps": int,
    "warmup": int, "ms_per_step": float, "higher_is_better": bool,
    "scaling": str, "dtype": str, "data": str, "config": dict,
}


def test_bench_json_contract():
    r = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "2",
         "--warmup", "1", "--device", "cpu", "--vocab", "2000", "--dim", "32",
         "--words-per-step", "10000", "--table-size", "5000"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    line = r.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for k, t in REQUIRED.items():
        assert k in d, f"missing {k}"
        assert isinstance(d[k], t), (k, type(d[k]))
    assert "vs_baseline" in d            # may be None
    assert d["value"] > 0
    assert d["scaling"] == "weak"
    assert d["higher_is_better"] is True
    assert "model" in d["config"] and "global_batch" in d["config"]
    assert "parallelism" in d["config"]
