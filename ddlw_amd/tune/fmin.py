"""``fmin`` + TPE suggester + Trials / LocalTrials.

Re-implements the hyperopt surface the reference exercises
(``Part 2 .../01_hyperopt_single_machine_model.py:221-243``,
``Part 2 .../02_hyperopt_distributed_model.py:352-365``):

- ``fmin(fn, space, algo=tpe.suggest, max_evals, trials)`` minimising the
  objective's ``{'loss': ..., 'status': STATUS_OK}`` dict (or a bare float);
- returns the best parameter dict with **indices for hp.choice** entries
  (hyperopt quirk);
- sequential ``Trials`` (in-process, the only mode compatible with nested
  distributed trials — reference :342-344);
- ``LocalTrials(parallelism=k)`` — the SparkTrials equivalent: k concurrent
  trial subprocesses, each pinned to its own GPU via HIP_VISIBLE_DEVICES
  (SURVEY.md §2.2 "Task parallelism" row).

The suggester is a Tree-of-Parzen-Estimators implementation (Bergstra et al.,
NeurIPS 2011) over the three primitives used: after ``n_startup`` random
trials, observations split into good/bad at the gamma-quantile of loss; each
numeric parameter is modelled by a Parzen mixture (truncated Gaussians around
observed values), categoricals by smoothed counts; candidates drawn from the
good density are ranked by g(x)/b(x).
"""
from __future__ import annotations

import math
import os
import traceback
from typing import Any, Callable, Dict, List, Optional

import numpy as np

from .space import sample_param, externalize

STATUS_OK = "ok"
STATUS_FAIL = "fail"


class Trials:
    """Sequential in-process trial store/executor."""

    parallelism = 1

    def __init__(self):
        self.trials: List[dict] = []

    @property
    def losses(self) -> List[Optional[float]]:
        return [t["result"].get("loss") for t in self.trials]

    @property
    def best_trial(self) -> dict:
        ok = [t for t in self.trials if t["result"].get("status") == STATUS_OK]
        if not ok:
            raise RuntimeError("no successful trials")
        return min(ok, key=lambda t: t["result"]["loss"])

    # executor interface ------------------------------------------------- #
    def run_batch(self, fn: Callable, batch: List[Dict[str, Any]]) -> List[dict]:
        return [_run_one(fn, params) for params in batch]


def _run_one(fn: Callable, params: Dict[str, Any]) -> dict:
    try:
        out = fn(params)
        if isinstance(out, dict):
            out.setdefault("status", STATUS_OK)
            return out
        return {"loss": float(out), "status": STATUS_OK}
    except Exception:
        return {"status": STATUS_FAIL, "error": traceback.format_exc()}


def _trial_worker(fn, params, env, q):
    os.environ.update(env)
    q.put(_run_one(fn, params))


class LocalTrials(Trials):
    """SparkTrials equivalent: run up to ``parallelism`` trials concurrently
    as subprocesses, each pinned to one GPU (HIP_VISIBLE_DEVICES rotation).
    """

    def __init__(self, parallelism: int = 4, gpus: Optional[List[int]] = None,
                 trial_timeout_s: Optional[float] = None):
        super().__init__()
        self.parallelism = max(1, parallelism)
        self.trial_timeout_s = trial_timeout_s  # None = unbounded (training)
        if gpus is None:
            try:
                import torch

                gpus = list(range(torch.cuda.device_count())) if torch.cuda.is_available() else []
            except Exception:
                gpus = []
        self.gpus = gpus

    def run_batch(self, fn: Callable, batch: List[Dict[str, Any]]) -> List[dict]:
        import multiprocessing as mp

        from ..core import tracking

        ctx = mp.get_context("spawn")
        results: List[Optional[dict]] = [None] * len(batch)
        active_run = tracking.active_run()
        base_env = {
            "DDLW_TRACKING_URI": tracking.get_tracking_uri(),
        }
        if active_run is not None:
            base_env["DDLW_PARENT_RUN_ID"] = active_run.run_id
        procs = []
        for i, params in enumerate(batch):
            env = dict(base_env)
            if self.gpus:
                gpu = self.gpus[i % len(self.gpus)]
                env["HIP_VISIBLE_DEVICES"] = str(gpu)
            q = ctx.SimpleQueue()
            p = ctx.Process(target=_trial_worker, args=(fn, params, env, q), daemon=False)
            p.start()
            procs.append((i, p, q))
        for i, p, q in procs:
            p.join(self.trial_timeout_s)
            if p.is_alive():
                # hung trial: kill it and record a failure instead of
                # blocking the whole sweep (SURVEY.md §5.3 discipline)
                p.terminate()
                p.join(5)
                results[i] = {
                    "status": STATUS_FAIL,
                    "error": f"trial timed out after {self.trial_timeout_s}s",
                }
                continue
            results[i] = (
                q.get()
                if not q.empty()
                else {"status": STATUS_FAIL, "error": f"trial exited {p.exitcode}"}
            )
        return results  # type: ignore[return-value]


# --------------------------------------------------------------------------- #
# TPE
# --------------------------------------------------------------------------- #


class _TPE:
    def __init__(self, n_startup_jobs: int = 5, gamma: float = 0.25, n_ei_candidates: int = 24):
        self.n_startup_jobs = n_startup_jobs
        self.gamma = gamma
        self.n_ei_candidates = n_ei_candidates

    # numeric Parzen estimator ------------------------------------------ #
    @staticmethod
    def _parzen_pdf(x: np.ndarray, obs: np.ndarray, low: float, high: float) -> np.ndarray:
        if len(obs) == 0:
            return np.full_like(x, 1.0 / (high - low))
        span = high - low
        sigma = max(span / max(len(obs), 1), 1e-6 * span)
        # mixture of truncated gaussians + uniform prior component
        comp = np.exp(-0.5 * ((x[:, None] - obs[None, :]) / sigma) ** 2) / (
            sigma * math.sqrt(2 * math.pi)
        )
        return (comp.sum(1) + 1.0 / span) / (len(obs) + 1)

    @staticmethod
    def _parzen_sample(rng, obs: np.ndarray, low: float, high: float, n: int) -> np.ndarray:
        span = high - low
        sigma = max(span / max(len(obs), 1), 1e-6 * span)
        out = np.empty(n)
        for i in range(n):
            if len(obs) == 0 or rng.random() < 1.0 / (len(obs) + 1):
                out[i] = rng.uniform(low, high)
            else:
                mu = obs[rng.integers(0, len(obs))]
                v = rng.normal(mu, sigma)
                out[i] = min(max(v, low), high)
        return out

    def suggest(self, space: Dict[str, Any], history: List[dict], rng) -> Dict[str, Any]:
        """Return one new internal-parameter dict. Trials with an empty
        result are PENDING (e.g. same-batch picks not yet run): they join
        the "bad" density so concurrent suggestions spread out instead of
        clustering (hyperopt's pending-trial treatment)."""
        done = [t for t in history if t["result"].get("status") == STATUS_OK]
        pending = [t for t in history if not t["result"]]
        if len(done) < self.n_startup_jobs:
            return {k: sample_param(spec, rng) for k, spec in space.items()}
        losses = np.array([t["result"]["loss"] for t in done])
        n_good = max(1, int(math.ceil(self.gamma * len(done))))
        order = np.argsort(losses, kind="stable")
        good = [done[i] for i in order[:n_good]]
        bad = [done[i] for i in order[n_good:]]

        out: Dict[str, Any] = {}
        for key, spec in space.items():
            gv = np.array([t["params"][key] for t in good], dtype=float)
            bv = np.array([t["params"][key] for t in bad + pending], dtype=float)
            if spec.kind == "choice":
                k = len(spec.options)
                gc = np.bincount(gv.astype(int), minlength=k) + 1.0
                bc = np.bincount(bv.astype(int), minlength=k) + 1.0
                score = (gc / gc.sum()) / (bc / bc.sum())
                probs = gc / gc.sum()
                cands = rng.choice(k, size=self.n_ei_candidates, p=probs)
                out[key] = int(cands[np.argmax(score[cands])])
                continue
            if spec.kind == "loguniform":
                low, high = spec.low, spec.high  # log-space bounds
                gv, bv = np.log(gv), np.log(bv)
                cands = self._parzen_sample(rng, gv, low, high, self.n_ei_candidates)
                g = self._parzen_pdf(cands, gv, low, high)
                b = self._parzen_pdf(cands, bv, low, high)
                out[key] = float(math.exp(cands[np.argmax(g / b)]))
                continue
            low, high = spec.low, spec.high
            cands = self._parzen_sample(rng, gv, low, high, self.n_ei_candidates)
            g = self._parzen_pdf(cands, gv, low, high)
            b = self._parzen_pdf(cands, bv, low, high)
            v = float(cands[np.argmax(g / b)])
            if spec.kind == "quniform":
                v = float(round(v / spec.q) * spec.q)
            out[key] = v
        return out


class _TPEModule:
    """Namespace so callers write ``algo=tpe.suggest`` like hyperopt."""

    suggest = "tpe"


tpe = _TPEModule()


class _RandModule:
    suggest = "rand"


rand = _RandModule()


def fmin(
    fn: Callable[[Dict[str, Any]], Any],
    space: Dict[str, Any],
    algo: str = "tpe",
    max_evals: int = 20,
    trials: Optional[Trials] = None,
    rstate: Optional[np.random.Generator] = None,
    verbose: bool = True,
) -> Dict[str, Any]:
    """Minimise ``fn`` over ``space``; returns the best internal param dict
    (choice params as indices — the hyperopt contract)."""
    algo_name = getattr(algo, "suggest", algo)
    if isinstance(algo_name, str) and algo_name not in ("tpe", "rand"):
        algo_name = "tpe"
    trials = trials if trials is not None else Trials()
    rng = rstate or np.random.default_rng(0)
    suggester = _TPE()
    while len(trials.trials) < max_evals:
        batch_n = min(trials.parallelism, max_evals - len(trials.trials))
        internal_batch = []
        pending = []  # same-batch picks, passed to TPE as result-less trials
        for _ in range(batch_n):
            if algo_name == "rand":
                internal = {k: sample_param(s, rng) for k, s in space.items()}
            else:
                internal = suggester.suggest(space, trials.trials + pending, rng)
                pending.append({"tid": -1, "params": internal, "result": {}})
            internal_batch.append(internal)
        external_batch = [
            {k: externalize(space[k], v) for k, v in internal.items()}
            for internal in internal_batch
        ]
        results = trials.run_batch(fn, external_batch)
        for internal, result in zip(internal_batch, results):
            trials.trials.append(
                {"tid": len(trials.trials), "params": internal, "result": result}
            )
            if verbose:
                loss = result.get("loss")
                if loss is None:
                    err = (result.get("error") or "").strip().splitlines()
                    detail = f" ({err[-1]})" if err else ""
                    print(
                        f"[fmin] trial {len(trials.trials)}/{max_evals} "
                        f"loss=FAIL{detail}",
                        flush=True,
                    )
                else:
                    print(
                        f"[fmin] trial {len(trials.trials)}/{max_evals} "
                        f"loss={loss}",
                        flush=True,
                    )
    best = trials.best_trial
    return dict(best["params"])
