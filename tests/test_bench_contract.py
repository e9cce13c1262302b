"""Driver contract: `python bench.py` emits ONE JSON line with the required
keys (the round driver parses this). Runs the shrunken CPU plumbing config."""

import json
import os
import subprocess
import sys

ROOT = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"),
         "--steps", "1", "--warmup", "0"],
        cwd=ROOT, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    j = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in j, key
    assert j["data"] == "synthetic"
    assert j["scaling"] == "weak"
    assert j["steps"] == 1
    for key in ("model", "global_batch", "seq_len", "parallelism"):
        assert key in j["config"], key
