"""Packaged predict-function — the ``mlflow.pyfunc`` equivalent.

Reference contract (``Part 2 .../03_pyfunc_distributed_inference.py``):

- ``class FlowerPyFunc(mlflow.pyfunc.PythonModel)`` with
  ``load_context(context)`` reading ``context.artifacts[name]`` paths and
  ``predict(context, pd.Series) -> np.ndarray`` (:157-212);
- ``mlflow.pyfunc.log_model('pyfunc_model', python_model=..., artifacts=
  {'name': 'runs:/.../path'})`` — artifact URIs resolved and *copied into* the
  packaged model so it is self-contained (:350-363);
- ``mlflow.pyfunc.load_model(uri).predict(series)`` (:446-448);
- ``mlflow.pyfunc.spark_udf(spark, uri, result_type='string')`` applied to a
  binary ``content`` column (:466-472) -> here :func:`predict_udf`, a fan-out
  over N local GPU worker processes, each loading the model once and
  predicting its row partition.

Packaged layout under the run's artifacts:

    <artifact_path>/
      MLmodel               # flavor: ddlw_pyfunc
      python_model.pkl      # cloudpickled PythonModel instance
      artifacts/<name>/...  # localized copies, keyed by name
"""
from __future__ import annotations

import json
import os
import shutil
from pathlib import Path
from typing import Dict, List, Optional, Sequence

import io

import cloudpickle
import numpy as np
import yaml

from ..core import tracking


def _subimport(name):
    import importlib

    importlib.import_module(name)
    import sys as _s

    return _s.modules[name]


class _DdlwPickler(cloudpickle.Pickler):
    """cloudpickle's submodule scan mis-detects ``torch.classes`` as a used
    submodule whenever a pickled function touches any ``.classes`` attribute
    (torch registers the lazy torch.classes module in sys.modules); that lazy
    module is unpicklable. Reduce it — and any torch._classes namespace — to
    a plain re-import."""

    def reducer_override(self, obj):
        mod = getattr(type(obj), "__module__", "")
        if mod == "torch._classes":
            name = getattr(obj, "__name__", None) or "torch.classes"
            if not name.startswith("torch"):
                name = "torch.classes"
            return (_subimport, (name,))
        return super().reducer_override(obj)


def _dumps(obj) -> bytes:
    buf = io.BytesIO()
    _DdlwPickler(buf).dump(obj)
    return buf.getvalue()


class PythonModel:
    """Subclass and implement ``load_context`` + ``predict``."""

    def load_context(self, context: "PyFuncContext") -> None:  # noqa: D401
        ...

    def predict(self, context: "PyFuncContext", model_input):
        raise NotImplementedError


class PyFuncContext:
    def __init__(self, artifacts: Dict[str, str]):
        self.artifacts = dict(artifacts)


def log_model(
    artifact_path: str,
    python_model: PythonModel,
    artifacts: Optional[Dict[str, str]] = None,
) -> str:
    """Package ``python_model`` + localized artifacts under the active run;
    returns its ``runs:/`` URI."""
    run = tracking.active_run()
    if run is None:
        run = tracking.start_run()
    root = Path(run.artifact_uri) / artifact_path
    adir = root / "artifacts"
    adir.mkdir(parents=True, exist_ok=True)
    localized: Dict[str, str] = {}
    for name, uri in (artifacts or {}).items():
        src = tracking.resolve_artifact_uri(uri)
        dst = adir / name
        if src.is_dir():
            shutil.copytree(src, dst, dirs_exist_ok=True)
        else:
            dst.mkdir(parents=True, exist_ok=True)
            shutil.copy2(src, dst / src.name)
            dst = dst / src.name
        localized[name] = str(dst.relative_to(root))
    (root / "python_model.pkl").write_bytes(_dumps(python_model))
    (root / "MLmodel").write_text(
        yaml.safe_dump(
            {
                "flavors": {"ddlw_pyfunc": {"loader": "ddlw_amd.infer.pyfunc"}},
                "artifacts": localized,
            }
        )
    )
    return f"runs:/{run.run_id}/{artifact_path}"


class PyFuncModel:
    def __init__(self, model: PythonModel, context: PyFuncContext):
        self._model = model
        self._context = context

    def predict(self, model_input):
        return self._model.predict(self._context, model_input)


def load_model(model_uri: str) -> PyFuncModel:
    root = tracking.resolve_artifact_uri(model_uri)
    meta = yaml.safe_load((root / "MLmodel").read_text())
    rel = meta.get("artifacts", {})
    context = PyFuncContext({k: str(root / v) for k, v in rel.items()})
    model: PythonModel = cloudpickle.loads((root / "python_model.pkl").read_bytes())
    model.load_context(context)
    return PyFuncModel(model, context)


# --------------------------------------------------------------------------- #
# distributed predict-UDF fan-out
# --------------------------------------------------------------------------- #


def _pool_worker(model_uri: str, env: Dict[str, str], task_q, result_q) -> None:
    os.environ.update(env)
    # Workers are non-daemonic (a packaged model may fork its own decode
    # pool — daemonic processes cannot have children), so guard against
    # outliving a crashed parent: poll the parent pid and exit if it changes
    # (orphaned -> reparented).
    import threading as _t

    ppid = os.getppid()

    def _watchdog():
        import time as _time

        while True:
            _time.sleep(2.0)
            if os.getppid() != ppid:
                os._exit(0)

    _t.Thread(target=_watchdog, daemon=True).start()
    try:
        m = load_model(model_uri)
        result_q.put(("ready", os.getpid(), None))
    except Exception:
        import traceback

        result_q.put(("err", os.getpid(), traceback.format_exc()))
        return
    while True:
        task = task_q.get()
        if task is None:
            return
        task_id, rows = task
        try:
            out = m.predict(rows)
            result_q.put(("ok", task_id, list(np.asarray(out).astype(str))))
        except Exception:
            import traceback

            result_q.put(("err", task_id, traceback.format_exc()))


class PredictUDF:
    """Persistent predict-UDF worker pool (the ``spark_udf`` contract,
    result_type='string'): each worker process loads the packaged model ONCE,
    pinned to its own GPU via HIP_VISIBLE_DEVICES, and serves row partitions
    until closed — like a Spark executor serving Arrow batches."""

    def __init__(self, model_uri: str, num_workers: Optional[int] = None,
                 gpus: Optional[List[int]] = None):
        import multiprocessing as mp

        if gpus is None:
            try:
                import torch

                gpus = list(range(torch.cuda.device_count())) if torch.cuda.is_available() else []
            except Exception:
                gpus = []
        if num_workers is None:
            num_workers = max(1, len(gpus)) if gpus else 2
        self.num_workers = num_workers
        ctx = mp.get_context("spawn")
        self._result_q = ctx.Queue()
        self._task_qs = []
        self._procs = []
        self._closed = False
        # non-daemon workers would block interpreter exit (multiprocessing
        # joins children at shutdown) if the caller forgets close(); our
        # atexit hook runs before mp's join and sends the sentinels
        import atexit

        atexit.register(self.close)
        base_env = {"DDLW_TRACKING_URI": tracking.get_tracking_uri()}
        for i in range(num_workers):
            env = dict(base_env)
            if gpus:
                env["HIP_VISIBLE_DEVICES"] = str(gpus[i % len(gpus)])
            tq = ctx.Queue()
            # daemon=False so the worker can fork a decode pool (see
            # _pool_worker's orphan watchdog for the cleanup guarantee)
            p = ctx.Process(
                target=_pool_worker, args=(model_uri, env, tq, self._result_q),
                daemon=False,
            )
            p.start()
            self._task_qs.append(tq)
            self._procs.append(p)
        ready = 0
        while ready < num_workers:
            import queue as _q

            try:
                status, _, payload = self._result_q.get(timeout=5.0)
            except _q.Empty:
                dead = [p.pid for p in self._procs if not p.is_alive()]
                if dead:
                    self.close()
                    raise RuntimeError(
                        f"predict worker(s) died during model load (pids {dead})"
                    )
                continue
            if status == "err":
                self.close()
                raise RuntimeError(f"predict worker failed to load model:\n{payload}")
            ready += 1

    def __call__(self, rows: Sequence) -> List[str]:
        rows = list(rows)
        if not rows:
            return []
        # drain results stranded by a previous failed call (worker died
        # mid-task -> we raised with results still in flight); stale task
        # ids must not be attributed to this call
        import queue as _q

        while True:
            try:
                self._result_q.get_nowait()
            except _q.Empty:
                break
        n = min(self.num_workers, len(rows))
        chunk = (len(rows) + n - 1) // n
        # ceil-chunking can produce trailing EMPTY parts on uneven splits
        # (e.g. 9 rows / 8 workers -> 2,2,2,2,1,0,0,0); drop them so no
        # worker is asked to predict([])
        parts = [p for p in (rows[i * chunk : (i + 1) * chunk] for i in range(n)) if p]
        n = len(parts)
        for i, part in enumerate(parts):
            self._task_qs[i].put((i, part))
        results: List[Optional[List[str]]] = [None] * n
        errs = []
        got = 0
        while got < n:
            # liveness-checked wait: a worker dying mid-task (OOM, kill)
            # must surface as an error, not a hang
            import queue as _q

            try:
                status, task_id, payload = self._result_q.get(timeout=5.0)
            except _q.Empty:
                dead = [p.pid for p in self._procs if not p.is_alive()]
                if dead:
                    raise RuntimeError(
                        f"predict worker(s) died mid-task (pids {dead})"
                    )
                continue
            got += 1
            if status == "ok":
                results[task_id] = payload
            else:
                errs.append(str(payload))
        if errs:
            raise RuntimeError("predict_udf worker failure:\n" + "\n".join(errs))
        out: List[str] = []
        for part in results:
            out.extend(part or [])
        return out

    def close(self) -> None:
        if self._closed:
            return
        self._closed = True
        for tq in self._task_qs:
            try:
                tq.put(None)
            except Exception:
                pass
        for p in self._procs:
            p.join(10)
            if p.is_alive():
                p.terminate()

    def __enter__(self) -> "PredictUDF":
        return self

    def __exit__(self, *exc) -> None:
        self.close()


def predict_udf(
    model_uri: str,
    num_workers: Optional[int] = None,
    gpus: Optional[List[int]] = None,
) -> PredictUDF:
    """Return a callable ``udf(rows) -> list[str]`` backed by a persistent
    worker pool; call ``.close()`` (or use as a context manager) when done."""
    return PredictUDF(model_uri, num_workers=num_workers, gpus=gpus)


def predict_table(
    model_uri: str,
    source,
    content_column: str = "content",
    output_column: str = "prediction",
    num_workers: Optional[int] = None,
    gpus: Optional[List[int]] = None,
    output_path: Optional[str] = None,
    limit: Optional[int] = None,
):
    """Table-in -> table-out batch inference: the reference's
    ``df.withColumn('prediction', loaded_model_udf('content'))`` contract
    (``Part 2 .../03_pyfunc_distributed_inference.py:466-472``).

    ``source`` is a pyarrow Table or a Parquet file/dataset-dir path. The
    ``content_column`` rows are fanned across the worker pool; the returned
    Table is the input (optionally ``limit``-ed, like the reference's
    ``.limit(1000)``) plus a string ``output_column``. When ``output_path``
    is given the result is also written there as Parquet.
    """
    import pyarrow as pa
    import pyarrow.parquet as pq

    if isinstance(source, (str, Path)):
        p = Path(source)
        if p.is_dir():
            parts = sorted(p.glob("*.parquet"))
            table = pa.concat_tables([pq.read_table(f) for f in parts])
        else:
            table = pq.read_table(p)
    else:
        table = source
    if limit is not None:
        table = table.slice(0, limit)
    rows = table.column(content_column).to_pylist()
    with PredictUDF(model_uri, num_workers=num_workers, gpus=gpus) as udf:
        preds = udf(rows)
    out = table.append_column(output_column, pa.array(preds, pa.string()))
    if output_path is not None:
        out_p = Path(output_path)
        out_p.parent.mkdir(parents=True, exist_ok=True)
        pq.write_table(out, out_p)
    return out
