"""Synthetic data generators (BASELINE configs run on synthetic data —
no network access for the TF-flowers dataset the reference uses)."""
from __future__ import annotations

import io
from pathlib import Path
from typing import List

import numpy as np
from PIL import Image


def make_synthetic_jpeg_tree(
    out_dir: str,
    num_classes: int = 5,
    images_per_class: int = 20,
    size: int = 64,
    seed: int = 0,
) -> Path:
    """Create ``out_dir/<class_name>/img_<i>.jpg`` — the same directory layout
    the reference's flowers dataset has (label = parent dir,
    ``Part 1 .../01_data_prep.py:125-130``)."""
    rng = np.random.default_rng(seed)
    out = Path(out_dir)
    class_names = [f"class_{c}" for c in range(num_classes)]
    for c, name in enumerate(class_names):
        d = out / name
        d.mkdir(parents=True, exist_ok=True)
        for i in range(images_per_class):
            # structured pattern so classes are learnable
            base = np.zeros((size, size, 3), np.float32)
            base[..., c % 3] = 128 + 64 * np.sin(
                np.linspace(0, 3 + c, size)[:, None] + np.linspace(0, 2, size)[None, :]
            )
            noise = rng.uniform(0, 96, (size, size, 3))
            arr = np.clip(base + noise, 0, 255).astype(np.uint8)
            Image.fromarray(arr).save(d / f"img_{i:04d}.jpg", quality=85)
    return out


def make_synthetic_dataset(
    n: int,
    img_height: int = 224,
    img_width: int = 224,
    channels: int = 3,
    num_classes: int = 5,
    seed: int = 0,
    jpeg: bool = False,
):
    """In-memory synthetic rows: (contents, labels). ``jpeg=True`` returns
    encoded JPEG bytes, else raw uint8 arrays."""
    rng = np.random.default_rng(seed)
    labels = rng.integers(0, num_classes, n).tolist()
    contents: List = []
    for i in range(n):
        arr = rng.integers(0, 256, (img_height, img_width, channels), dtype=np.uint8)
        if jpeg:
            buf = io.BytesIO()
            Image.fromarray(arr).save(buf, format="JPEG", quality=85)
            contents.append(buf.getvalue())
        else:
            contents.append(arr)
    return contents, labels
