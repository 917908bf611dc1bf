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


def test_conv_dispatch_table_schema():
    """The measured router table must stay well-formed: every entry maps
    fwd/dgrad/wgrad to 'hip' or 'stock' (a corrupted table would silently
    route everything to the library)."""
    import json
    from pathlib import Path

    table = json.loads(
        (REPO / "ddlw_amd" / "ops" / "conv_dispatch.json").read_text())
    assert len(table) >= 20  # the 22 ResNet-50 body shapes
    for key, ent in table.items():
        assert set(ent) == {"fwd", "dgrad", "wgrad"}, key
        assert all(v in ("hip", "stock") for v in ent.values()), key
    # fwd must be fully hip (measured winner on all 22 routes)
    assert all(e["fwd"] == "hip" for e in table.values())
