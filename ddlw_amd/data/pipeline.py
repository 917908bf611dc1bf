"""Bronze/silver data pipeline: JPEG tree -> Parquet tables.

Re-implements ``Part 1 - Distributed Training/01_data_prep.py`` without Spark:

- bronze: recursive ``*.jpg`` scan with optional sampling
  (reference :61-66: binaryFile reader, ``recursiveFileLookup``, ``sample(0.5)``)
  -> columns ``path, modificationTime, length, content`` (:50-53),
  written *uncompressed* (binary JPEG does not recompress, :92);
- silver: + ``label`` = parent directory name (pandas-UDF semantics, :125-130);
- deterministic 90/10 split seeded 42 (:162);
- ``label_to_idx`` built from the *sorted distinct train labels* (:179-182 —
  quirk 6 in SURVEY.md §2.6: the map intentionally comes from the train split
  only; a val-only class raises KeyError exactly as the reference would);
- silver_train / silver_val with ``label_idx`` (:187-222).

Tables are Parquet datasets under ``<database_dir>/<table>/``; row-group size
is chosen for per-rank sharding by the streaming loader (SURVEY.md §2.5
Petastorm row).
"""
from __future__ import annotations

import random
from pathlib import Path
from typing import Dict, List, Optional

import pyarrow as pa
import pyarrow.parquet as pq

from ..core.config import current_setup

BRONZE_SCHEMA = pa.schema(
    [
        ("path", pa.string()),
        ("modificationTime", pa.timestamp("ms")),
        ("length", pa.int64()),
        ("content", pa.binary()),
    ]
)


def table_path(table: str, database_name: Optional[str] = None) -> Path:
    s = current_setup()
    db = database_name or s.database_name
    return Path(s.root) / "warehouse" / db / table


def _write_table(tbl: pa.Table, table: str, database_name: Optional[str], row_group_rows: int) -> Path:
    out = table_path(table, database_name)
    out.mkdir(parents=True, exist_ok=True)
    f = out / "part-00000.parquet"
    # compression NONE for binary image content (reference :92)
    pq.write_table(tbl, f, row_group_size=row_group_rows, compression="NONE")
    return out


def read_table(table: str, database_name: Optional[str] = None, columns: Optional[List[str]] = None) -> pa.Table:
    return pq.read_table(str(table_path(table, database_name)), columns=columns)


def build_tables(
    img_dir: str,
    database_name: Optional[str] = None,
    sample_fraction: float = 0.5,
    train_fraction: float = 0.9,
    seed: int = 42,
    row_group_rows: int = 64,
) -> Dict[str, int]:
    """Scan ``img_dir`` recursively for .jpg files and build
    bronze / silver / silver_train / silver_val. Returns row counts."""
    img_dir = Path(img_dir)
    files = sorted(p for p in img_dir.rglob("*.jpg"))
    rng = random.Random(seed)
    if sample_fraction < 1.0:
        files = [p for p in files if rng.random() < sample_fraction]

    paths, mtimes, lengths, contents = [], [], [], []
    for p in files:
        st = p.stat()
        paths.append(str(p))
        mtimes.append(int(st.st_mtime * 1000))
        lengths.append(st.st_size)
        contents.append(p.read_bytes())
    bronze = pa.Table.from_arrays(
        [
            pa.array(paths),
            pa.array(mtimes, pa.timestamp("ms")),
            pa.array(lengths, pa.int64()),
            pa.array(contents, pa.binary()),
        ],
        schema=BRONZE_SCHEMA,
    )
    _write_table(bronze, "bronze", database_name, row_group_rows)

    # silver: label = parent dir name (reference :125-130)
    labels = [Path(p).parent.name for p in paths]
    silver = bronze.append_column("label", pa.array(labels))
    _write_table(silver, "silver", database_name, row_group_rows)

    # deterministic 90/10 split (reference :162 randomSplit seed=42)
    split_rng = random.Random(seed)
    is_train = [split_rng.random() < train_fraction for _ in range(len(files))]
    train_idx = [i for i, t in enumerate(is_train) if t]
    val_idx = [i for i, t in enumerate(is_train) if not t]

    # label map from SORTED DISTINCT TRAIN labels only (reference :179-182)
    train_labels = sorted({labels[i] for i in train_idx})
    label_to_idx = {lab: i for i, lab in enumerate(train_labels)}

    def _subset(idxs: List[int], name: str) -> pa.Table:
        sub = silver.take(pa.array(idxs, pa.int64()))
        # KeyError on a val-only class is the reference's behaviour (quirk 6)
        li = [label_to_idx[labels[i]] for i in idxs]
        sub = sub.append_column("label_idx", pa.array(li, pa.int64()))
        _write_table(sub, name, database_name, row_group_rows)
        return sub

    tr = _subset(train_idx, "silver_train")
    va = _subset(val_idx, "silver_val")

    import json

    meta = table_path("silver_train", database_name).parent / "label_to_idx.json"
    meta.write_text(json.dumps(label_to_idx, indent=2))

    return {
        "bronze": bronze.num_rows,
        "silver": silver.num_rows,
        "silver_train": tr.num_rows,
        "silver_val": va.num_rows,
        "num_classes": len(label_to_idx),
    }
