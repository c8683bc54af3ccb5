"""bench.py contract smoke on CPU (the real numbers come from the GPU run;
this checks the JSON contract the driver depends on)."""
import json
import subprocess
import sys


def test_bench_json_contract_cpu():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "1",
         "--device", "cpu", "--batch-size", "16"],
        capture_output=True, text=True, timeout=300, check=True)
    line = [l for l in out.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    r = json.loads(line)
    for key in ["metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"]:
        assert key in r, key
    assert r["scaling"] == "weak"
    assert r["data"] == "synthetic"
    assert r["n_gpus"] == 1
    assert r["value"] > 0
    assert r["config"]["parallelism"] == "dp1"
