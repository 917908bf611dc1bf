"""TPE/fmin tests: space sampling bounds, convergence, hp.choice index
contract, LocalTrials parallel execution (SURVEY.md §4 item 1 + §2.6 #4)."""
import math

import numpy as np
import pytest

from ddlw_amd.tune import LocalTrials, STATUS_OK, Trials, fmin, hp, tpe
from ddlw_amd.tune.space import externalize, sample_param


def test_space_sampling_bounds():
    rng = np.random.default_rng(0)
    u = hp.uniform("d", 0.1, 0.9)
    lu = hp.loguniform("lr", math.log(1e-5), math.log(1.0))
    c = hp.choice("opt", ["Adam", "Adadelta"])
    for _ in range(200):
        v = sample_param(u, rng)
        assert 0.1 <= v <= 0.9
        v = sample_param(lu, rng)
        assert 1e-5 <= v <= 1.0
        i = sample_param(c, rng)
        assert i in (0, 1)
        assert externalize(c, i) in ("Adam", "Adadelta")


def test_fmin_converges_quadratic():
    space = {"x": hp.uniform("x", -10.0, 10.0)}
    best = fmin(lambda p: (p["x"] - 3.0) ** 2, space, algo=tpe.suggest, max_evals=40, verbose=False)
    assert abs(best["x"] - 3.0) < 1.5


def test_fmin_beats_random_on_average():
    space = {"x": hp.uniform("x", 0.0, 1.0), "y": hp.uniform("y", 0.0, 1.0)}

    def obj(p):
        return (p["x"] - 0.2) ** 2 + (p["y"] - 0.7) ** 2

    rng = np.random.default_rng(1)
    tl, rl = [], []
    for s in range(3):
        t = Trials()
        fmin(obj, space, algo=tpe.suggest, max_evals=30, trials=t,
             rstate=np.random.default_rng(s), verbose=False)
        tl.append(min(x for x in t.losses if x is not None))
        rnd = min(
            obj({"x": rng.uniform(), "y": rng.uniform()}) for _ in range(30)
        )
        rl.append(rnd)
    assert np.mean(tl) <= np.mean(rl) * 1.5  # TPE at least competitive


def test_choice_returns_index():
    space = {
        "optimizer": hp.choice("optimizer", ["Adam", "Adadelta"]),
        "lr": hp.loguniform("lr", math.log(1e-4), math.log(1e-1)),
    }

    def obj(p):
        assert p["optimizer"] in ("Adam", "Adadelta")  # objective sees the value
        return {"loss": p["lr"], "status": STATUS_OK}

    best = fmin(obj, space, max_evals=8, verbose=False)
    # hyperopt quirk: fmin returns the INDEX
    assert best["optimizer"] in (0, 1)


def _parallel_obj(p):
    return {"loss": (p["x"] - 0.5) ** 2, "status": STATUS_OK}


def test_local_trials_parallel(ddlw_home):
    space = {"x": hp.uniform("x", 0.0, 1.0)}
    trials = LocalTrials(parallelism=3, gpus=[])
    best = fmin(_parallel_obj, space, max_evals=6, trials=trials, verbose=False)
    assert len(trials.trials) == 6
    assert 0.0 <= best["x"] <= 1.0


def _failing_obj(p):
    raise ValueError("bad trial")


def test_failed_trials_recorded(ddlw_home):
    trials = Trials()
    with pytest.raises(RuntimeError):
        fmin(_failing_obj, {"x": hp.uniform("x", 0, 1)}, max_evals=3, trials=trials, verbose=False)
    assert all(t["result"]["status"] == "fail" for t in trials.trials)


def _sleepy_objective(params):
    import time

    time.sleep(120)
    return 0.0


def test_local_trials_timeout(ddlw_home):
    """A hung trial is killed and recorded as FAIL, not awaited forever."""
    import time

    from ddlw_amd.tune.fmin import STATUS_FAIL, LocalTrials

    t0 = time.time()
    trials = LocalTrials(parallelism=1, gpus=[], trial_timeout_s=5)
    res = trials.run_batch(_sleepy_objective, [{"x": 1.0}])
    assert res[0]["status"] == STATUS_FAIL and "timed out" in res[0]["error"]
    assert time.time() - t0 < 60
