"""Search-space DSL — the ``hyperopt.hp`` subset the reference uses.

Reference spaces (``Part 2 .../01_hyperopt_single_machine_model.py:194-198``,
``Part 2 .../02_hyperopt_distributed_model.py:322-326``):
``hp.choice``, ``hp.uniform``, ``hp.loguniform`` (loguniform(low, high) samples
``exp(U(low, high))``, hyperopt semantics).

Quirk preserved: ``fmin`` returns the *index* for
``hp.choice`` parameters, exactly like hyperopt.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Any, Sequence


@dataclass
class Choice:
    label: str
    options: Sequence[Any]

    kind = "choice"


@dataclass
class Uniform:
    label: str
    low: float
    high: float

    kind = "uniform"


@dataclass
class LogUniform:
    label: str
    low: float  # log-space bounds, hyperopt-style: sample = exp(U(low, high))
    high: float

    kind = "loguniform"


@dataclass
class QUniform:
    label: str
    low: float
    high: float
    q: float

    kind = "quniform"


class _HP:
    @staticmethod
    def choice(label: str, options: Sequence[Any]) -> Choice:
        return Choice(label, list(options))

    @staticmethod
    def uniform(label: str, low: float, high: float) -> Uniform:
        return Uniform(label, float(low), float(high))

    @staticmethod
    def loguniform(label: str, low: float, high: float) -> LogUniform:
        return LogUniform(label, float(low), float(high))

    @staticmethod
    def quniform(label: str, low: float, high: float, q: float) -> QUniform:
        return QUniform(label, float(low), float(high), float(q))


hp = _HP()


def sample_param(spec, rng) -> Any:
    """Draw one *internal* value: index for choice, float otherwise."""
    if spec.kind == "choice":
        return int(rng.integers(0, len(spec.options)))
    if spec.kind == "uniform":
        return float(rng.uniform(spec.low, spec.high))
    if spec.kind == "loguniform":
        return float(math.exp(rng.uniform(spec.low, spec.high)))
    if spec.kind == "quniform":
        v = rng.uniform(spec.low, spec.high)
        return float(round(v / spec.q) * spec.q)
    raise TypeError(f"unknown spec {spec}")


def externalize(spec, internal) -> Any:
    """Internal value -> the value the objective sees (choice -> option)."""
    if spec.kind == "choice":
        return spec.options[int(internal)]
    return internal
