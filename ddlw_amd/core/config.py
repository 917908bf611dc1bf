"""Config module — the ``00_setup`` + ``DataCfg`` equivalents.

The reference has three config tiers: ``00_setup.py`` globals
(user -> database name, host/token; ``Part 1 - Distributed Training/00_setup.py:3-17``),
per-notebook UPPERCASE constants, and one typed dataclass
(``DataCfg``, ``Part 2 .../03_pyfunc_distributed_inference.py:85-94``).
Here everything is a dataclass; ``setup()`` plays the role of ``%run ./00_setup``.
"""
from __future__ import annotations

import getpass
import os
import re
from dataclasses import dataclass, field, asdict
from pathlib import Path
from typing import Optional, Tuple


def _default_root() -> Path:
    return Path(os.environ.get("DDLW_HOME", Path.home() / ".ddlw_amd"))


@dataclass
class SetupCfg:
    """Equivalent of ``00_setup.py``: derives a per-user database name and the
    tracking location (reference: ``.../00_setup.py:3-17`` — user e-mail ->
    ``{name}_db``; DATABRICKS_HOST/TOKEN -> here a plain filesystem URI)."""

    user: str = field(default_factory=getpass.getuser)
    root: Path = field(default_factory=_default_root)

    @property
    def my_name(self) -> str:
        return re.sub(r"\W", "_", self.user.split("@")[0])

    @property
    def database_name(self) -> str:
        return f"{self.my_name}_db"

    @property
    def database_dir(self) -> Path:
        return Path(self.root) / "warehouse" / self.database_name

    @property
    def tracking_uri(self) -> str:
        return str(Path(self.root) / "mlruns")


_SETUP: Optional[SetupCfg] = None


def setup(root: Optional[str] = None, user: Optional[str] = None) -> SetupCfg:
    """Initialise global config (idempotent), like ``%run ./00_setup``."""
    global _SETUP
    kw = {}
    if root is not None:
        kw["root"] = Path(root)
        # export so worker subprocesses (trials, ranks, UDF workers) resolve
        # the same data root (mirrors DDLW_TRACKING_URI propagation)
        os.environ["DDLW_HOME"] = str(root)
    if user is not None:
        kw["user"] = user
    if _SETUP is None or kw:
        _SETUP = SetupCfg(**kw)
    return _SETUP


def current_setup() -> SetupCfg:
    return _SETUP if _SETUP is not None else setup()


@dataclass
class DataCfg:
    """Typed data config (reference: ``Part 2 .../03_pyfunc_distributed_inference.py:85-94``)."""

    img_height: int = 224
    img_width: int = 224
    img_channels: int = 3
    num_classes: int = 5
    batch_size: int = 128
    database_name: str = ""
    train_table: str = "silver_train"
    val_table: str = "silver_val"
    cache_dir: str = ""

    def __post_init__(self):
        s = current_setup()
        if not self.database_name:
            self.database_name = s.database_name
        if not self.cache_dir:
            self.cache_dir = str(Path(s.root) / "cache")

    @property
    def img_params(self) -> dict:
        return {
            "img_height": self.img_height,
            "img_width": self.img_width,
            "img_channels": self.img_channels,
            "num_classes": self.num_classes,
        }


@dataclass
class TrainCfg:
    """Per-run training constants (the UPPERCASE notebook cells, e.g.
    ``Part 1 .../02_model_training_single_node.py:41-46``)."""

    batch_size: int = 32
    epochs: int = 3
    learning_rate: float = 1e-3
    dropout: float = 0.5
    optimizer: str = "Adam"
    dtype: str = "bf16"
    seed: int = 42

    def as_dict(self) -> dict:
        return asdict(self)


@dataclass
class ImageShape:
    height: int = 224
    width: int = 224
    channels: int = 3

    @property
    def hwc(self) -> Tuple[int, int, int]:
        return (self.height, self.width, self.channels)
