"""bench.py driver contract: one JSON line on stdout with the agreed
fields (the round-end harness parses exactly this)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract_cpu():
    r = subprocess.run(
        [sys.executable, "bench.py", "--batch", "8", "--steps", "2",
         "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    d = json.loads(lines[0])
    for key in ["metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"]:
        assert key in d, key
    assert d["unit"] == "images/sec"
    assert d["scaling"] == "weak"
    assert d["higher_is_better"] is True
    assert d["data"] == "synthetic"
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["config"]["model"] == "resnet18"
    assert d["config"]["global_batch"] == 8
    assert d["value"] > 0
