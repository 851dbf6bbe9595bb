"""bench.py output obeys the driver contract."""
import json
import subprocess
import sys


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "8", "--distinct", "2", "--nodes", "20"],
        capture_output=True, text=True, timeout=600)
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in d, key
    assert d["metric"] == "episodes/sec"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert abs(d["vs_baseline"] - d["value"] / 4.0) < 1e-6
    assert "global_batch" in d["config"] and "parallelism" in d["config"]
