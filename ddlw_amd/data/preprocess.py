"""Image preprocessing — the reference's ``preprocess()`` centralised.

Reference (duplicated across 5 notebooks, SURVEY.md §2.6 quirk 7;
canonical copy ``Part 1 .../02_model_training_single_node.py:119-126``):
``decode_jpeg(content) -> resize(h, w) -> MobileNetV2 preprocess_input``
where preprocess_input scales uint8 to [-1, 1] (x/127.5 - 1).

Two decode paths, as in the reference: the main (tf.image) path and the PIL
path used inside the pyfunc (``Part 2 .../03_pyfunc_distributed_inference.py:214-234``)
including the str->bytes ``ast.literal_eval`` workaround for string-typed
content columns.
"""
from __future__ import annotations

import ast
import io
from typing import Iterable

import numpy as np
import torch
from PIL import Image


def _to_bytes(content) -> bytes:
    if isinstance(content, (bytes, bytearray)):
        return bytes(content)
    if isinstance(content, str):
        # Spark-UDF string-typed column workaround
        # (reference: Part 2 .../03_pyfunc_distributed_inference.py:228-229)
        return ast.literal_eval(content)
    return bytes(content)


def decode_jpeg(content) -> np.ndarray:
    """bytes -> uint8 HWC RGB array."""
    img = Image.open(io.BytesIO(_to_bytes(content))).convert("RGB")
    return np.asarray(img)


def preprocess_pil(content, img_height: int = 224, img_width: int = 224) -> np.ndarray:
    """PIL decode+resize path (pyfunc variant). Returns float32 HWC in [-1,1]."""
    img = Image.open(io.BytesIO(_to_bytes(content))).convert("RGB")
    img = img.resize((img_width, img_height), Image.BILINEAR)
    arr = np.asarray(img, dtype=np.float32)
    return arr / 127.5 - 1.0


def preprocess_bytes(content, img_height: int = 224, img_width: int = 224) -> torch.Tensor:
    """Main path: decode -> bilinear resize -> [-1,1]. Returns CHW float32
    (torch layout; the HWC->CHW transpose is the only deviation from the
    reference's tensor layout and is internal)."""
    arr = np.array(decode_jpeg(content))  # owned, writable copy
    t = torch.from_numpy(arr).permute(2, 0, 1).float().unsqueeze(0)
    t = torch.nn.functional.interpolate(
        t, size=(img_height, img_width), mode="bilinear", align_corners=False
    )
    return (t / 127.5 - 1.0).squeeze(0)


def preprocess_batch(
    contents: Iterable, img_height: int = 224, img_width: int = 224
) -> torch.Tensor:
    """Decode+preprocess a batch of JPEG byte strings -> B x C x H x W float32."""
    return torch.stack([preprocess_bytes(c, img_height, img_width) for c in contents])


def normalize_uint8(x: torch.Tensor) -> torch.Tensor:
    """uint8 tensor -> float in [-1,1] (the preprocess_input transform alone,
    for pre-decoded synthetic pipelines)."""
    return x.to(torch.float32) / 127.5 - 1.0
