"""Tracking store + registry tests (MLflow-layout golden tests)."""
import json

import pytest
import yaml

from ddlw_amd.core import tracking


def test_run_lifecycle_and_layout(ddlw_home):
    tracking.set_experiment("exp1")
    with tracking.start_run(run_name="r1") as run:
        run.log_param("lr", 0.001)
        run.log_metric("loss", 1.5, step=0)
        run.log_metric("loss", 1.0, step=1)
        run.log_dict({"a": 1}, "cfg.json")
        rid = run.run_id
    # on-disk MLflow layout
    d = run.dir
    assert (d / "meta.yaml").exists()
    assert (d / "params" / "lr").read_text() == "0.001"
    lines = (d / "metrics" / "loss").read_text().strip().splitlines()
    assert len(lines) == 2 and lines[1].split()[1] == "1.0"
    assert json.loads((d / "artifacts" / "cfg.json").read_text()) == {"a": 1}
    meta = yaml.safe_load((d / "meta.yaml").read_text())
    assert meta["status"] == "FINISHED" and meta["run_id"] == rid


def test_reattach_by_run_id(ddlw_home):
    tracking.set_experiment("exp1")
    run = tracking.start_run(run_name="parent")
    rid = run.run_id
    tracking.end_run()
    # worker-style re-attach (reference P1/03:361-373)
    r2 = tracking.start_run(run_id=rid)
    r2.log_metric("val_acc", 0.9)
    tracking.end_run()
    assert tracking.get_run(rid).metrics()["val_acc"] == 0.9


def test_nested_runs_and_search(ddlw_home):
    tracking.set_experiment("hpo")
    with tracking.start_run(run_name="parent") as parent:
        for i in range(3):
            with tracking.start_run(run_name=f"c{i}", nested=True) as child:
                child.log_metric("accuracy", 0.5 + 0.1 * i)
    df = tracking.search_runs(
        filter_string=f'tags.mlflow.parentRunId = "{parent.run_id}"',
        order_by=["metrics.accuracy DESC"],
    )
    assert len(df) == 3
    assert df.iloc[0]["metrics.accuracy"] == pytest.approx(0.7)
    assert df.iloc[-1]["metrics.accuracy"] == pytest.approx(0.5)


def test_registry_stage_transitions(ddlw_home):
    tracking.set_experiment("reg")
    with tracking.start_run() as run:
        run.log_text("weights", "model/weights.txt")
        uri = f"runs:/{run.run_id}/model"
    mv = tracking.register_model(uri, "flowers")
    assert mv["version"] == 1
    tracking.transition_model_version_stage("flowers", 1, "Production")
    path = tracking.resolve_artifact_uri("models:/flowers/production")
    assert (path / "weights.txt").read_text() == "weights"


def test_filter_parse_errors(ddlw_home):
    with pytest.raises(ValueError):
        tracking._parse_filter("malformed ~~ clause")


def test_search_runs_missing_order_key(ddlw_home):
    """Reference quirk 3: ordering by a metric some runs
    never logged must not crash — absent keys sort last, present ones win."""
    from ddlw_amd.core import tracking

    tracking.set_experiment("order_edge")
    with tracking.start_run(run_name="a") as r:
        r.log_metric("val_loss", 0.5)        # no 'accuracy' at all
    with tracking.start_run(run_name="b") as r:
        r.log_metric("accuracy", 0.9)
    df = tracking.search_runs(order_by=["metrics.accuracy DESC"])
    assert len(df) == 2
    assert df.iloc[0]["metrics.accuracy"] == 0.9  # the run that HAS the key


def test_search_runs_like_operator(ddlw_home):
    from ddlw_amd.core import tracking

    tracking.set_experiment("like_edge")
    with tracking.start_run(run_name="trial-1") as r:
        r.set_tag("kind", "hpo")
    with tracking.start_run(run_name="other") as r:
        r.set_tag("kind", "manual")
    df = tracking.search_runs(filter_string="tags.kind = 'hpo'")
    assert len(df) == 1
