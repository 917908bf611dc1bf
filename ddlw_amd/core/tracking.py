"""File-backed experiment tracking with the MLflow on-disk layout.

Re-implements, dependency-free, the MLflow surface the reference exercises
(SURVEY.md §2.5 "MLflow tracking/models/registry/pyfunc" rows):

- fluent runs: ``start_run(run_name=..., run_id=..., nested=...)`` including
  re-attach by id from a worker process
  (``Part 1 .../03_model_training_distributed.py:361-373``) and nested child
  runs under a parent (``Part 2 .../02_hyperopt_distributed_model.py:241-260``);
- ``log_param/log_params/log_metric/log_metrics/log_dict/log_text/set_tag``;
- ``search_runs(filter_string=..., order_by=...)`` -> pandas DataFrame
  (``Part 2 .../01_hyperopt_single_machine_model.py:253-262``);
- model logging under ``runs:/{run_id}/model`` and reload
  (``Part 1 .../03_model_training_distributed.py:373,438-439``);
- a model registry: ``register_model`` + ``transition_model_version_stage`` +
  ``models:/{name}/production`` URIs
  (``Part 2 .../01_hyperopt_single_machine_model.py:279-298``).

On-disk layout (MLflow `file:` store compatible):

    <root>/<experiment_id>/meta.yaml
    <root>/<experiment_id>/<run_id>/meta.yaml
    <root>/<experiment_id>/<run_id>/params/<key>          # one value per file
    <root>/<experiment_id>/<run_id>/metrics/<key>         # "ts value step" lines
    <root>/<experiment_id>/<run_id>/tags/<key>
    <root>/<experiment_id>/<run_id>/artifacts/...
    <root>/models/<name>/version-<N>/meta.yaml            # registry

Worker processes on other ranks/GPUs append to the same store through the
``DDLW_TRACKING_URI`` env var (the reference ships DATABRICKS_HOST/TOKEN to
workers for the same purpose, ``.../03_model_training_distributed.py:286-288``).
"""
from __future__ import annotations

import json
import os
import re
import time
import uuid
import threading
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Dict, List, Optional

import yaml

_lock = threading.RLock()

# --------------------------------------------------------------------------- #
# store location
# --------------------------------------------------------------------------- #

_tracking_uri: Optional[str] = None


def set_tracking_uri(uri: str) -> None:
    global _tracking_uri
    _tracking_uri = str(uri)
    os.environ["DDLW_TRACKING_URI"] = _tracking_uri


def get_tracking_uri() -> str:
    global _tracking_uri
    if _tracking_uri is None:
        env = os.environ.get("DDLW_TRACKING_URI")
        if env:
            _tracking_uri = env
        else:
            from .config import current_setup

            _tracking_uri = current_setup().tracking_uri
    return _tracking_uri


def _root() -> Path:
    p = Path(get_tracking_uri())
    p.mkdir(parents=True, exist_ok=True)
    return p


def _now_ms() -> int:
    return int(time.time() * 1000)


# --------------------------------------------------------------------------- #
# experiments
# --------------------------------------------------------------------------- #

DEFAULT_EXPERIMENT = "0"


def _experiment_dir(experiment_id: str) -> Path:
    return _root() / experiment_id


def create_experiment(name: str) -> str:
    with _lock:
        existing = get_experiment_by_name(name)
        if existing is not None:
            return existing["experiment_id"]
        ids = [int(d.name) for d in _root().iterdir() if d.is_dir() and d.name.isdigit()]
        eid = str(max(ids) + 1 if ids else 0)
        d = _experiment_dir(eid)
        d.mkdir(parents=True, exist_ok=True)
        meta = {
            "experiment_id": eid,
            "name": name,
            "artifact_location": str(d),
            "lifecycle_stage": "active",
            "creation_time": _now_ms(),
        }
        (d / "meta.yaml").write_text(yaml.safe_dump(meta))
        return eid


def get_experiment_by_name(name: str) -> Optional[dict]:
    for d in sorted(_root().iterdir()) if _root().exists() else []:
        mf = d / "meta.yaml"
        if d.is_dir() and mf.exists():
            meta = yaml.safe_load(mf.read_text())
            if isinstance(meta, dict) and meta.get("name") == name:
                return meta
    return None


_active_experiment_id: Optional[str] = None


def set_experiment(name: str) -> str:
    global _active_experiment_id
    _active_experiment_id = create_experiment(name)
    # worker processes (trials, ranks, UDF workers) inherit the experiment
    os.environ["DDLW_EXPERIMENT_ID"] = _active_experiment_id
    return _active_experiment_id


def _current_experiment_id() -> str:
    global _active_experiment_id
    if _active_experiment_id is None:
        env = os.environ.get("DDLW_EXPERIMENT_ID")
        _active_experiment_id = env if env else create_experiment("Default")
    return _active_experiment_id


# --------------------------------------------------------------------------- #
# runs
# --------------------------------------------------------------------------- #


def _find_run_dir(run_id: str) -> Path:
    for d in _root().iterdir():
        if d.is_dir():
            rd = d / run_id
            if rd.is_dir() and (rd / "meta.yaml").exists():
                return rd
    raise KeyError(f"run_id {run_id!r} not found under {_root()}")


class Run:
    """Handle for an active (or re-attached) run."""

    def __init__(self, run_id: str, experiment_id: str, run_dir: Path):
        self.run_id = run_id
        self.experiment_id = experiment_id
        self.dir = run_dir

    # mlflow.ActiveRun compat
    @property
    def info(self):
        return self

    @property
    def artifact_uri(self) -> str:
        return str(self.dir / "artifacts")

    def __enter__(self) -> "Run":
        return self

    def __exit__(self, exc_type, exc, tb) -> None:
        end_run("FAILED" if exc_type else "FINISHED")

    # ------------------------------------------------------------------ #
    def _meta(self) -> dict:
        return yaml.safe_load((self.dir / "meta.yaml").read_text())

    def _write_meta(self, meta: dict) -> None:
        (self.dir / "meta.yaml").write_text(yaml.safe_dump(meta))

    def log_param(self, key: str, value: Any) -> None:
        p = self.dir / "params"
        p.mkdir(exist_ok=True)
        (p / _safe_key(key)).write_text(str(value))

    def log_params(self, params: Dict[str, Any]) -> None:
        for k, v in params.items():
            self.log_param(k, v)

    def log_metric(self, key: str, value: float, step: int = 0) -> None:
        p = self.dir / "metrics"
        p.mkdir(exist_ok=True)
        with open(p / _safe_key(key), "a") as f:
            f.write(f"{_now_ms()} {float(value)} {int(step)}\n")

    def log_metrics(self, metrics: Dict[str, float], step: int = 0) -> None:
        for k, v in metrics.items():
            self.log_metric(k, v, step)

    def set_tag(self, key: str, value: Any) -> None:
        p = self.dir / "tags"
        p.mkdir(exist_ok=True)
        (p / _safe_key(key)).write_text(str(value))

    def log_dict(self, d: dict, artifact_file: str) -> None:
        path = self.dir / "artifacts" / artifact_file
        path.parent.mkdir(parents=True, exist_ok=True)
        if artifact_file.endswith((".yaml", ".yml")):
            path.write_text(yaml.safe_dump(d))
        else:
            path.write_text(json.dumps(d, indent=2))

    def log_text(self, text: str, artifact_file: str) -> None:
        path = self.dir / "artifacts" / artifact_file
        path.parent.mkdir(parents=True, exist_ok=True)
        path.write_text(text)

    def log_artifact(self, local_path: str, artifact_path: str = "") -> None:
        import shutil

        src = Path(local_path)
        dst = self.dir / "artifacts" / artifact_path / src.name
        dst.parent.mkdir(parents=True, exist_ok=True)
        if src.is_dir():
            shutil.copytree(src, dst, dirs_exist_ok=True)
        else:
            shutil.copy2(src, dst)

    # read side ---------------------------------------------------------- #
    def params(self) -> Dict[str, str]:
        p = self.dir / "params"
        return {f.name: f.read_text() for f in p.iterdir()} if p.exists() else {}

    def metrics(self) -> Dict[str, float]:
        """Latest value per metric key."""
        p = self.dir / "metrics"
        out: Dict[str, float] = {}
        if p.exists():
            for f in p.iterdir():
                lines = f.read_text().strip().splitlines()
                if lines:
                    out[f.name] = float(lines[-1].split()[1])
        return out

    def metric_history(self, key: str) -> List[tuple]:
        f = self.dir / "metrics" / _safe_key(key)
        if not f.exists():
            return []
        out = []
        for line in f.read_text().strip().splitlines():
            ts, v, s = line.split()
            out.append((int(ts), float(v), int(s)))
        return out

    def tags(self) -> Dict[str, str]:
        p = self.dir / "tags"
        return {f.name: f.read_text() for f in p.iterdir()} if p.exists() else {}


def _safe_key(key: str) -> str:
    return re.sub(r"[^\w.\-]", "_", key)


_run_stack: List[Run] = []


def start_run(
    run_name: Optional[str] = None,
    run_id: Optional[str] = None,
    nested: bool = False,
    experiment_id: Optional[str] = None,
    tags: Optional[Dict[str, str]] = None,
) -> Run:
    """Start (or re-attach to) a run. Mirrors ``mlflow.start_run`` as used at
    ``Part 1 .../03_model_training_distributed.py:363`` (re-attach by id) and
    ``Part 2 .../02_hyperopt_distributed_model.py:244-247`` (nested)."""
    global _run_stack
    with _lock:
        if run_id is not None:
            rd = _find_run_dir(run_id)
            run = Run(run_id, rd.parent.name, rd)
            _run_stack.append(run)
            return run
        if _run_stack and not nested:
            raise RuntimeError(
                "Run already active; use nested=True or end_run() first "
                "(mlflow fluent-API contract)"
            )
        eid = experiment_id or _current_experiment_id()
        rid = uuid.uuid4().hex
        rd = _experiment_dir(eid) / rid
        (rd / "artifacts").mkdir(parents=True, exist_ok=True)
        meta = {
            "run_id": rid,
            "run_uuid": rid,
            "run_name": run_name or rid[:8],
            "experiment_id": eid,
            "status": "RUNNING",
            "start_time": _now_ms(),
            "end_time": None,
            "artifact_uri": str(rd / "artifacts"),
            "lifecycle_stage": "active",
        }
        (rd / "meta.yaml").write_text(yaml.safe_dump(meta))
        run = Run(rid, eid, rd)
        if run_name:
            run.set_tag("mlflow.runName", run_name)
        if nested and _run_stack:
            run.set_tag("mlflow.parentRunId", _run_stack[-1].run_id)
        parent_env = os.environ.get("DDLW_PARENT_RUN_ID")
        if nested and not _run_stack and parent_env:
            # worker-side nesting via env (reference: MLFLOW_PARENT_RUN_ID at
            # Part 2 .../02_hyperopt_distributed_model.py:241-247)
            run.set_tag("mlflow.parentRunId", parent_env)
        for k, v in (tags or {}).items():
            run.set_tag(k, v)
        _run_stack.append(run)
        return run


def active_run() -> Optional[Run]:
    return _run_stack[-1] if _run_stack else None


def end_run(status: str = "FINISHED") -> None:
    with _lock:
        if not _run_stack:
            return
        run = _run_stack.pop()
        meta = run._meta()
        if meta.get("status") == "RUNNING":
            meta["status"] = status
            meta["end_time"] = _now_ms()
            run._write_meta(meta)


def get_run(run_id: str) -> Run:
    rd = _find_run_dir(run_id)
    return Run(run_id, rd.parent.name, rd)


# convenience module-level logging against the active run ------------------- #


def _require_active() -> Run:
    r = active_run()
    if r is None:
        r = start_run()
    return r


def log_param(key: str, value: Any) -> None:
    _require_active().log_param(key, value)


def log_params(params: Dict[str, Any]) -> None:
    _require_active().log_params(params)


def log_metric(key: str, value: float, step: int = 0) -> None:
    _require_active().log_metric(key, value, step)


def log_metrics(metrics: Dict[str, float], step: int = 0) -> None:
    _require_active().log_metrics(metrics, step)


def log_dict(d: dict, artifact_file: str) -> None:
    _require_active().log_dict(d, artifact_file)


def set_tag(key: str, value: Any) -> None:
    _require_active().set_tag(key, value)


# --------------------------------------------------------------------------- #
# search_runs
# --------------------------------------------------------------------------- #

_FILTER_RE = re.compile(
    r"""\s*(?P<field>[\w.]+|tags\.`[^`]+`|tags\."[^"]+")\s*"""
    r"""(?P<op>=|!=|>=|<=|>|<|LIKE)\s*"""
    r"""(?P<val>'[^']*'|"[^"]*"|[-\w.]+)\s*""",
    re.IGNORECASE,
)


def _parse_filter(filter_string: str) -> List[tuple]:
    """Parse a (subset of the) MLflow search filter grammar:
    ``tags.mlflow.parentRunId = "xyz" and metrics.accuracy > 0.5``."""
    clauses = []
    if not filter_string:
        return clauses
    for part in re.split(r"\s+and\s+", filter_string, flags=re.IGNORECASE):
        m = _FILTER_RE.fullmatch(part)
        if not m:
            raise ValueError(f"unsupported filter clause: {part!r}")
        fieldname = m.group("field").replace("`", "").replace('"', "")
        val = m.group("val").strip("'\"")
        clauses.append((fieldname, m.group("op"), val))
    return clauses


def _run_field(run: Run, meta: dict, fieldname: str):
    if fieldname.startswith("tags."):
        return run.tags().get(fieldname[5:])
    if fieldname.startswith("params."):
        return run.params().get(fieldname[7:])
    if fieldname.startswith("metrics."):
        return run.metrics().get(fieldname[8:])
    if fieldname.startswith("attributes."):
        fieldname = fieldname[11:]
    return meta.get(fieldname)


def search_runs(
    experiment_ids: Optional[List[str]] = None,
    filter_string: str = "",
    order_by: Optional[List[str]] = None,
    max_results: int = 1000,
    experiment_names: Optional[List[str]] = None,
):
    """Return a pandas DataFrame of runs, MLflow-style: columns ``run_id``,
    ``experiment_id``, ``status``, ``params.*``, ``metrics.*``, ``tags.*``.
    (Reference usage: ``Part 2 .../01_hyperopt_single_machine_model.py:253-262``.)
    """
    import pandas as pd

    if experiment_names:
        experiment_ids = []
        for n in experiment_names:
            e = get_experiment_by_name(n)
            if e:
                experiment_ids.append(e["experiment_id"])
    if experiment_ids is None:
        experiment_ids = [_current_experiment_id()]
    clauses = _parse_filter(filter_string)
    rows = []
    for eid in experiment_ids:
        ed = _experiment_dir(eid)
        if not ed.exists():
            continue
        for rd in ed.iterdir():
            if not rd.is_dir() or not (rd / "meta.yaml").exists():
                continue
            meta = yaml.safe_load((rd / "meta.yaml").read_text())
            run = Run(meta["run_id"], eid, rd)
            ok = True
            for fieldname, op, val in clauses:
                actual = _run_field(run, meta, fieldname)
                if actual is None:
                    ok = False
                    break
                try:
                    a, b = float(actual), float(val)
                except (TypeError, ValueError):
                    a, b = str(actual), str(val)
                if op == "=" and not a == b:
                    ok = False
                elif op == "!=" and not a != b:
                    ok = False
                elif op == ">" and not a > b:
                    ok = False
                elif op == "<" and not a < b:
                    ok = False
                elif op == ">=" and not a >= b:
                    ok = False
                elif op == "<=" and not a <= b:
                    ok = False
                elif op.upper() == "LIKE" and str(val).replace("%", "") not in str(actual):
                    ok = False
                if not ok:
                    break
            if not ok:
                continue
            row: Dict[str, Any] = {
                "run_id": meta["run_id"],
                "experiment_id": eid,
                "status": meta.get("status"),
                "start_time": meta.get("start_time"),
                "end_time": meta.get("end_time"),
                "artifact_uri": meta.get("artifact_uri"),
            }
            for k, v in run.params().items():
                row[f"params.{k}"] = v
            for k, v in run.metrics().items():
                row[f"metrics.{k}"] = v
            for k, v in run.tags().items():
                row[f"tags.{k}"] = v
            rows.append(row)
    df = pd.DataFrame(rows)
    if order_by and len(df):
        for spec in reversed(order_by):
            parts = spec.rsplit(" ", 1)
            col = parts[0].replace("`", "")
            asc = len(parts) == 1 or parts[1].upper() != "DESC"
            if col in df.columns:
                df = df.sort_values(col, ascending=asc, kind="stable")
        df = df.reset_index(drop=True)
    return df.head(max_results)


# --------------------------------------------------------------------------- #
# artifact URI resolution
# --------------------------------------------------------------------------- #


def resolve_artifact_uri(uri: str) -> Path:
    """Resolve ``runs:/<run_id>/<path>`` and ``models:/<name>/<stage-or-ver>``
    URIs to local paths (the reference loads models by both forms:
    ``Part 1 .../03_model_training_distributed.py:438``,
    ``Part 2 .../01_hyperopt_single_machine_model.py:298``)."""
    if uri.startswith("runs:/"):
        rest = uri[len("runs:/") :]
        run_id, _, sub = rest.partition("/")
        return _find_run_dir(run_id) / "artifacts" / sub
    if uri.startswith("models:/"):
        rest = uri[len("models:/") :]
        name, _, sel = rest.partition("/")
        mv = _resolve_model_version(name, sel)
        return Path(mv["source"])
    return Path(uri)


# --------------------------------------------------------------------------- #
# model registry
# --------------------------------------------------------------------------- #

STAGES = ("None", "Staging", "Production", "Archived")


def _registry_dir() -> Path:
    d = _root() / "models"
    d.mkdir(parents=True, exist_ok=True)
    return d


def register_model(model_uri: str, name: str) -> dict:
    """Register the artifacts at ``model_uri`` as a new version of ``name``
    (reference: ``Part 2 .../01_hyperopt_single_machine_model.py:279-283``)."""
    with _lock:
        src = resolve_artifact_uri(model_uri)
        if not src.exists():
            raise FileNotFoundError(f"model_uri {model_uri} -> {src} does not exist")
        md = _registry_dir() / name
        md.mkdir(exist_ok=True)
        versions = [
            int(v.name.split("-")[1])
            for v in md.iterdir()
            if v.is_dir() and v.name.startswith("version-")
        ]
        ver = max(versions) + 1 if versions else 1
        vd = md / f"version-{ver}"
        vd.mkdir()
        meta = {
            "name": name,
            "version": ver,
            "source": str(src),
            "run_id": _run_id_from_uri(model_uri),
            "current_stage": "None",
            "creation_timestamp": _now_ms(),
        }
        (vd / "meta.yaml").write_text(yaml.safe_dump(meta))
        return meta


def _run_id_from_uri(uri: str) -> Optional[str]:
    if uri.startswith("runs:/"):
        return uri[len("runs:/") :].partition("/")[0]
    return None


def get_model_versions(name: str) -> List[dict]:
    md = _registry_dir() / name
    if not md.exists():
        return []
    out = []
    for vd in sorted(md.iterdir()):
        mf = vd / "meta.yaml"
        if mf.exists():
            out.append(yaml.safe_load(mf.read_text()))
    return out


def transition_model_version_stage(name: str, version: int, stage: str) -> dict:
    """(Reference: ``Part 2 .../01_hyperopt_single_machine_model.py:288-293``.)"""
    if stage not in STAGES:
        raise ValueError(f"stage must be one of {STAGES}")
    with _lock:
        vd = _registry_dir() / name / f"version-{version}"
        mf = vd / "meta.yaml"
        meta = yaml.safe_load(mf.read_text())
        meta["current_stage"] = stage
        meta["last_updated_timestamp"] = _now_ms()
        mf.write_text(yaml.safe_dump(meta))
        return meta


def _resolve_model_version(name: str, selector: str) -> dict:
    versions = get_model_versions(name)
    if not versions:
        raise KeyError(f"no registered model named {name!r}")
    if selector.isdigit():
        for v in versions:
            if v["version"] == int(selector):
                return v
        raise KeyError(f"{name} has no version {selector}")
    stage = selector.capitalize()
    staged = [v for v in versions if v.get("current_stage") == stage]
    if not staged:
        raise KeyError(f"{name} has no version in stage {stage!r}")
    return max(staged, key=lambda v: v["version"])
