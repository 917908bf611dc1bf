"""The driver depends on bench.py's exact contract: single JSON line with the
required keys, N=1 default, finishes quickly. Validate on CPU."""
import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_bench_json_contract(tmp_path):
    env = dict(os.environ)
    env["DDLW_HOME"] = str(tmp_path)
    res = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=900, env=env, cwd=str(REPO),
    )
    assert res.returncode == 0, res.stderr
    line = res.stdout.strip().splitlines()[-1]
    out = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in out, key
    assert out["n_gpus"] == 1
    assert out["steps"] == 2 and out["warmup"] == 1
    assert out["scaling"] == "weak"
    assert out["higher_is_better"] is True
    assert "global_batch" in out["config"] and "parallelism" in out["config"]
