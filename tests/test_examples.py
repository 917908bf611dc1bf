"""End-to-end integration tests: run every example flow on CPU, small scale
(SURVEY.md §4 item 4 — notebook-equivalent scripts per reference file)."""
import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


def _run(script, *args, home, timeout=600):
    env = dict(os.environ)
    env["DDLW_HOME"] = str(home)
    env.pop("DDLW_TRACKING_URI", None)
    env["PYTHONPATH"] = str(REPO)
    res = subprocess.run(
        [sys.executable, str(REPO / "examples" / script), "--root", str(home), *args],
        capture_output=True, text=True, timeout=timeout, env=env, cwd=str(REPO),
    )
    assert res.returncode == 0, f"{script} failed:\n{res.stdout}\n{res.stderr}"
    return res.stdout


@pytest.fixture(scope="module")
def prepared_home(tmp_path_factory):
    home = tmp_path_factory.mktemp("ddlw_e2e")
    out = _run("01_data_prep.py", "--synthetic", "--images-per-class", "12",
               "--sample-fraction", "1.0", home=home)
    assert "silver_train" in out
    return home


def test_02_single_node_training(prepared_home):
    out = _run("02_model_training_single_node.py", "--epochs", "1", home=prepared_home)
    assert "model logged under runs:/" in out


def test_03_distributed_training(prepared_home):
    out = _run("03_model_training_distributed.py", "--np", "2", "--epochs", "1",
               home=prepared_home)
    assert "smoke run" in out and "val_loss=" in out and "reloaded:" in out


def test_05_hyperopt_single_machine(prepared_home):
    out = _run("05_hyperopt_single_machine.py", "--max-evals", "4",
               "--parallelism", "2", home=prepared_home)
    assert "production model loaded" in out


def test_06_hyperopt_distributed(prepared_home):
    out = _run("06_hyperopt_distributed.py", "--max-evals", "2", "--np", "2",
               home=prepared_home)
    assert "checkpoints under" in out and "best child run" in out


def test_07_pyfunc_inference(prepared_home):
    out = _run("07_pyfunc_distributed_inference.py", "--workers", "2",
               home=prepared_home)
    assert "fanned out" in out
