"""Data pipeline tests: bronze/silver tables, label map determinism, split
(SURVEY.md §4 unit-test plan)."""
import json

import pyarrow.parquet as pq
import pytest

from ddlw_amd.data import build_tables, make_synthetic_jpeg_tree, read_table, table_path
from ddlw_amd.data.preprocess import preprocess_bytes, preprocess_pil


@pytest.fixture()
def jpeg_tree(ddlw_home, tmp_path):
    return make_synthetic_jpeg_tree(tmp_path / "imgs", num_classes=3, images_per_class=8, size=32)


def test_build_tables(jpeg_tree, ddlw_home):
    counts = build_tables(jpeg_tree, sample_fraction=1.0, row_group_rows=4)
    assert counts["bronze"] == 24
    assert counts["silver_train"] + counts["silver_val"] == 24
    assert counts["num_classes"] == 3
    silver = read_table("silver")
    assert set(silver.column_names) >= {"path", "length", "content", "label"}
    labels = set(silver.column("label").to_pylist())
    assert labels == {"class_0", "class_1", "class_2"}
    # label map is sorted-distinct-train (reference P1/01:179-182)
    m = json.loads((table_path("silver_train").parent / "label_to_idx.json").read_text())
    assert list(m.keys()) == sorted(m.keys())
    assert sorted(m.values()) == list(range(len(m)))


def test_split_deterministic(jpeg_tree, ddlw_home):
    c1 = build_tables(jpeg_tree, sample_fraction=1.0, seed=42)
    t1 = read_table("silver_train").column("path").to_pylist()
    c2 = build_tables(jpeg_tree, sample_fraction=1.0, seed=42)
    t2 = read_table("silver_train").column("path").to_pylist()
    assert t1 == t2 and c1 == c2


def test_row_groups_sized(jpeg_tree, ddlw_home):
    build_tables(jpeg_tree, sample_fraction=1.0, row_group_rows=4)
    f = next(table_path("silver_train").glob("*.parquet"))
    md = pq.ParquetFile(f).metadata
    assert md.num_row_groups >= 2


def test_preprocess_range_and_parity(jpeg_tree, ddlw_home):
    jpg = next(jpeg_tree.rglob("*.jpg")).read_bytes()
    t = preprocess_bytes(jpg, 24, 24)
    assert t.shape == (3, 24, 24)
    assert t.min() >= -1.0 and t.max() <= 1.0
    # PIL path parity: same scaling, same shape (HWC), close values
    p = preprocess_pil(jpg, 24, 24)
    assert p.shape == (24, 24, 3)
    assert p.min() >= -1.0 and p.max() <= 1.0
    # str-typed content workaround path (reference P2/03:228-229)
    p2 = preprocess_pil(str(jpg), 24, 24)
    assert (p2 == p).all()


def test_setup_root_propagates_to_env(tmp_path, monkeypatch):
    """setup(root=...) must export DDLW_HOME so trial/rank subprocesses
    resolve the same warehouse (regression: GPU-box HPO trials failed with
    FileNotFoundError on the default root)."""
    import os

    import ddlw_amd.core.config as config

    config._SETUP = None
    config.setup(root=str(tmp_path / "r1"))
    assert os.environ["DDLW_HOME"] == str(tmp_path / "r1")
    config._SETUP = None
