"""bench.py output contract (the driver parses this JSON line)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--model", "lenet",
         "--steps", "2", "--warmup", "1", "--no-graph"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in rec, key
    assert rec["metric"] == "images/sec (whole node)"
    assert rec["n_gpus"] == 1 and rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"
    assert rec["data"] == "synthetic"
    assert rec["value"] > 0
    cfg = rec["config"]
    for key in ("model", "global_batch", "input", "parallelism"):
        assert key in cfg, key
    assert cfg["parallelism"] == "dp1"
