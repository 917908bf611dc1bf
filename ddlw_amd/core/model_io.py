"""Model artifact save/load — the ``mlflow.keras.log_model/load_model`` equivalent.

The reference persists the trained Keras model under the run's ``model/``
artifact dir and reloads it by URI
(``Part 1 .../03_model_training_distributed.py:373,438-439``;
``Part 2 .../01_hyperopt_single_machine_model.py:298``). Here a model artifact is:

    model/
      MLmodel            # yaml: flavor info + builder spec
      state_dict.pt      # torch state dict (always saved on CPU)
      builder.json       # {"builder": "<registered name>", "kwargs": {...}}

Rebuildable models register a builder via ``ddlw_amd.models.register_builder``;
``load_model`` reconstructs the module and loads weights.
"""
from __future__ import annotations

import json
from pathlib import Path
from typing import Callable, Dict, Optional

import torch
import yaml

from . import tracking

_BUILDERS: Dict[str, Callable[..., "torch.nn.Module"]] = {}


def register_builder(name: str, fn: Callable[..., "torch.nn.Module"]) -> None:
    _BUILDERS[name] = fn


def get_builder(name: str) -> Callable[..., "torch.nn.Module"]:
    if name not in _BUILDERS:
        # importing models registers the stock builders
        from .. import models  # noqa: F401
    return _BUILDERS[name]


def save_model(model: torch.nn.Module, path: Path) -> None:
    path = Path(path)
    path.mkdir(parents=True, exist_ok=True)
    spec = getattr(model, "_ddlw_builder_spec", None)
    sd = {k: v.detach().cpu() for k, v in model.state_dict().items()}
    torch.save(sd, path / "state_dict.pt")
    builder = {"builder": spec[0], "kwargs": spec[1]} if spec else None
    (path / "builder.json").write_text(json.dumps(builder, indent=2))
    (path / "MLmodel").write_text(
        yaml.safe_dump(
            {
                "flavors": {
                    "ddlw_torch": {
                        "framework": "pytorch-rocm",
                        "builder": builder["builder"] if builder else None,
                    }
                },
                "model_format": "ddlw_state_dict_v1",
            }
        )
    )


def log_model(model: torch.nn.Module, artifact_path: str = "model") -> str:
    """Save ``model`` under the active run's artifacts; returns ``runs:/`` URI."""
    run = tracking.active_run()
    if run is None:
        run = tracking.start_run()
    dst = Path(run.artifact_uri) / artifact_path
    save_model(model, dst)
    return f"runs:/{run.run_id}/{artifact_path}"


def load_model(model_uri: str, map_location: str = "cpu") -> torch.nn.Module:
    """Load a model from ``runs:/``, ``models:/`` or a plain path."""
    path = tracking.resolve_artifact_uri(model_uri)
    builder_file = path / "builder.json"
    if not builder_file.exists():
        raise FileNotFoundError(f"{model_uri} -> {path} is not a ddlw model dir")
    spec = json.loads(builder_file.read_text())
    if spec is None:
        raise ValueError(
            f"model at {model_uri} was saved without a registered builder; "
            "load its state_dict.pt manually"
        )
    model = get_builder(spec["builder"])(**spec["kwargs"])
    sd = torch.load(path / "state_dict.pt", map_location=map_location, weights_only=True)
    model.load_state_dict(sd)
    return model


def tag_model(model: torch.nn.Module, builder: str, kwargs: Optional[dict] = None) -> torch.nn.Module:
    """Attach the rebuild spec used by ``save_model``."""
    model._ddlw_builder_spec = (builder, dict(kwargs or {}))
    return model
