#!/usr/bin/env python3
"""Data prep: JPEG tree -> bronze/silver/train/val Parquet tables.

Equivalent of ``Part 1 - Distributed Training/01_data_prep.py`` (binaryFile
scan -> bronze -> label-from-path silver -> seeded 90/10 split -> indexed
train/val tables). With --synthetic it first generates a synthetic JPEG tree
(no network in this environment for the TF-flowers download).
"""
import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import argparse

from ddlw_amd.core import setup
from ddlw_amd.data import build_tables, make_synthetic_jpeg_tree, read_table


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--img-dir", default=None)
    ap.add_argument("--synthetic", action="store_true")
    ap.add_argument("--num-classes", type=int, default=5)
    ap.add_argument("--images-per-class", type=int, default=40)
    ap.add_argument("--sample-fraction", type=float, default=0.5)
    ap.add_argument("--root", default=None)
    args = ap.parse_args()

    s = setup(root=args.root)
    img_dir = args.img_dir
    if args.synthetic or img_dir is None:
        img_dir = str(s.root / "flower_photos")
        make_synthetic_jpeg_tree(
            img_dir, num_classes=args.num_classes,
            images_per_class=args.images_per_class, size=64,
        )
    counts = build_tables(img_dir, sample_fraction=args.sample_fraction)
    print(f"database: {s.database_name}")
    for k, v in counts.items():
        print(f"  {k}: {v}")
    silver = read_table("silver_train", columns=["path", "label", "label_idx"])
    print(silver.slice(0, 5).to_pandas())


if __name__ == "__main__":
    main()
