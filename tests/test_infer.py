"""pyfunc tests: packaged-model round trip, artifact localization, predict
UDF single-node vs fanned-out equality."""
import json

import numpy as np
import torch

from ddlw_amd.core import tracking
from ddlw_amd.core.model_io import log_model
from ddlw_amd.data.synthetic import make_synthetic_dataset
from ddlw_amd.infer import PythonModel, load_model, log_model as log_pyfunc, predict_udf
from ddlw_amd.models import build_small_cnn

CLASSES = ["daisy", "dandelion", "rose", "sunflower", "tulip"]


class FlowerPyFunc(PythonModel):
    """Mirror of the reference's FlowerPyFunc (P2/03:157-234): load img params
    + model from packaged artifacts; predict = preprocess -> argmax -> label."""

    def load_context(self, context):
        from ddlw_amd.core.model_io import load_model as load_torch

        with open(context.artifacts["img_params_dict_path"]) as f:
            self.img_params = json.load(f)
        self.model = load_torch(context.artifacts["torch_model_path"])
        self.model.eval()

    def predict(self, context, model_input):
        from ddlw_amd.data.preprocess import preprocess_pil

        h, w = self.img_params["img_height"], self.img_params["img_width"]
        arrs = np.stack([preprocess_pil(c, h, w) for c in model_input])
        x = torch.from_numpy(arrs).permute(0, 3, 1, 2).float()
        with torch.no_grad():
            logits = self.model(x)
        idx = logits.argmax(-1).numpy()
        return np.take(CLASSES, idx)


def _package(ddlw_home):
    tracking.set_experiment("pyfunc")
    with tracking.start_run() as run:
        m = build_small_cnn(16, 16, num_classes=5)
        model_uri = log_model(m, "model")
        run.log_dict(
            {"img_height": 16, "img_width": 16, "img_channels": 3, "num_classes": 5},
            "img_params_dict.json",
        )
        uri = log_pyfunc(
            "pyfunc_model",
            FlowerPyFunc(),
            artifacts={
                "img_params_dict_path": f"runs:/{run.run_id}/img_params_dict.json",
                "torch_model_path": model_uri,
            },
        )
    return uri


def test_pyfunc_roundtrip(ddlw_home):
    uri = _package(ddlw_home)
    contents, _ = make_synthetic_dataset(6, 16, 16, num_classes=5, jpeg=True)
    m = load_model(uri)
    preds = m.predict(contents)
    assert len(preds) == 6
    assert all(p in CLASSES for p in preds)
    # str-typed content (the spark_udf string-column workaround) matches bytes
    preds_str = m.predict([str(c) for c in contents])
    assert list(preds) == list(preds_str)


def test_predict_udf_matches_single_node(ddlw_home):
    uri = _package(ddlw_home)
    contents, _ = make_synthetic_dataset(10, 16, 16, num_classes=5, seed=2, jpeg=True)
    single = list(load_model(uri).predict(contents))
    with predict_udf(uri, num_workers=3, gpus=[]) as udf:
        fanned = udf(contents)
    assert fanned == [str(s) for s in single]


class _DyingModel(PythonModel):
    """Predict kills the worker process — the pool must raise, not hang."""

    def load_context(self, context):
        pass

    def predict(self, context, rows):
        import os as _os

        _os._exit(41)


def test_predict_udf_detects_dead_worker(ddlw_home):
    import time

    import pytest

    from ddlw_amd.infer.pyfunc import log_model, predict_udf

    tracking.set_experiment("udf_death")
    with tracking.start_run():
        uri = log_model("killer", _DyingModel())
    t0 = time.time()
    with predict_udf(uri, num_workers=1) as udf:
        with pytest.raises(RuntimeError, match="died"):
            udf([b"row"])
    assert time.time() - t0 < 60


def test_predict_udf_uneven_split_no_empty_parts(ddlw_home):
    """5 rows on 4 workers must not dispatch empty chunks
    (models like np.stack-based ones raise on empty input)."""
    uri = _package(ddlw_home)
    contents, _ = make_synthetic_dataset(5, 16, 16, num_classes=5, seed=3, jpeg=True)
    single = list(load_model(uri).predict(contents))
    with predict_udf(uri, num_workers=4, gpus=[]) as udf:
        fanned = udf(contents)
        assert fanned == [str(s) for s in single]
        # reuse the pool: a second call must be unaffected
        assert udf(contents[:2]) == [str(s) for s in single[:2]]


def _times_two(content):
    import numpy as np

    return np.frombuffer(content, dtype=np.uint8).reshape(2, 2) * 2


def test_parallel_decoder_matches_serial():
    from ddlw_amd.data.decode import ParallelDecoder

    rows = [bytes([i, i + 1, i + 2, i + 3]) for i in range(0, 40, 4)]
    with ParallelDecoder(_times_two, workers=3, chunk_size=3) as dec:
        out = dec.map(rows)
        # pool reuse across calls
        out2 = dec.map(rows[:4])
    serial = ParallelDecoder(_times_two, workers=0).map(rows)
    assert torch.equal(out, serial)
    assert torch.equal(out2, serial[:4])


class _ForkingModel(PythonModel):
    """Model whose predict forks a decode pool — requires non-daemonic
    UDF workers (the inference decode path)."""

    def load_context(self, context):
        pass

    def predict(self, context, rows):
        from ddlw_amd.data.decode import ParallelDecoder

        with ParallelDecoder(_times_two, workers=2, chunk_size=2) as dec:
            out = dec.map(rows)
        return [str(int(t.sum())) for t in out]


def test_predict_udf_worker_can_fork_decode_pool(ddlw_home):
    from ddlw_amd.infer.pyfunc import log_model, predict_udf

    tracking.set_experiment("udf_fork")
    with tracking.start_run():
        uri = log_model("forking", _ForkingModel())
    rows = [bytes([i, i + 1, i + 2, i + 3]) for i in range(0, 24, 4)]
    expect = [str(int(_times_two(r).sum())) for r in rows]
    with predict_udf(uri, num_workers=2, gpus=[]) as udf:
        assert udf(rows) == expect


def test_predict_table_adds_prediction_column(ddlw_home, tmp_path):
    """Table-in/table-out contract (reference P2/03:466-472): input table +
    string prediction column; parquet round-trip; limit like .limit(1000)."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    from ddlw_amd.infer import load_model, predict_table

    uri = _package(ddlw_home)
    contents, labels = make_synthetic_dataset(12, 16, 16, num_classes=5, seed=7, jpeg=True)
    tbl = pa.table({"content": pa.array(contents, pa.binary()),
                    "label_idx": labels})
    out_pq = tmp_path / "pred" / "out.parquet"
    out = predict_table(uri, tbl, num_workers=2, gpus=[], limit=10,
                        output_path=str(out_pq))
    assert out.num_rows == 10
    assert out.column_names == ["content", "label_idx", "prediction"]
    single = [str(s) for s in load_model(uri).predict(contents[:10])]
    assert out.column("prediction").to_pylist() == single
    assert pq.read_table(out_pq).column("prediction").to_pylist() == single


def test_parallel_decoder_imap_order_and_content():
    from ddlw_amd.data.decode import ParallelDecoder

    rows = [bytes([i, i + 1, i + 2, i + 3]) for i in range(0, 80, 4)]
    serial = ParallelDecoder(_times_two, workers=0).map(rows)
    with ParallelDecoder(_times_two, workers=3, chunk_size=4) as dec:
        got = []
        for chunk in dec.imap(rows):
            got.append(chunk.clone())  # view only valid within iteration
    out = torch.cat(got)
    assert torch.equal(out, serial)
    # uneven tail
    with ParallelDecoder(_times_two, workers=2, chunk_size=3) as dec:
        sizes = [c.shape[0] for c in dec.imap(rows[:7])]
    assert sizes == [3, 3, 1]


def test_parallel_decoder_survives_abandoned_imap():
    """Breaking out of imap mid-stream must not corrupt the next call
    (stale in-flight results are drained)."""
    from ddlw_amd.data.decode import ParallelDecoder

    rows = [bytes([i, i + 1, i + 2, i + 3]) for i in range(0, 80, 4)]
    serial = ParallelDecoder(_times_two, workers=0).map(rows)
    with ParallelDecoder(_times_two, workers=3, chunk_size=2) as dec:
        for i, chunk in enumerate(dec.imap(rows)):
            if i == 2:
                break  # abandon mid-stream with tasks in flight
        out = dec.map(rows)  # pool reuse must still be correct
    assert torch.equal(out, serial)


def test_predict_table_from_parquet_dir(ddlw_home, tmp_path):
    """predict_table accepts a Parquet dataset DIRECTORY (bronze/silver
    layout) as the input side of the withColumn contract."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    from ddlw_amd.infer import predict_table

    uri = _package(ddlw_home)
    contents, labels = make_synthetic_dataset(8, 16, 16, num_classes=5, seed=8, jpeg=True)
    d = tmp_path / "ds"
    d.mkdir()
    t = pa.table({"content": pa.array(contents, pa.binary()), "label_idx": labels})
    pq.write_table(t.slice(0, 4), d / "part-0.parquet")
    pq.write_table(t.slice(4, 4), d / "part-1.parquet")
    out = predict_table(uri, str(d), num_workers=2, gpus=[])
    assert out.num_rows == 8 and "prediction" in out.column_names
