"""Local multi-process launcher — the ``HorovodRunner`` contract (C5).

Reference (``Part 1 .../03_model_training_distributed.py:258-263, 385-417``):
``HorovodRunner(np=N).run(fn, **kwargs)`` pickles ``fn``+closure, launches N
worker processes, runs ``fn`` under an initialised communicator in each, and
returns rank 0's return value. ``np=-1`` runs ``fn`` once *in the current
process* (the built-in smoke-test mode, :385-394).

MI355X-native shape: no Spark barrier mode, no mpirun — a ``spawn``-context
fork of N local processes, env rendezvous on 127.0.0.1 (RANK / WORLD_SIZE /
MASTER_ADDR / MASTER_PORT / LOCAL_RANK), one process per GPU over RCCL.
Failure detection: any worker dying aborts the whole job.
"""
from __future__ import annotations

import multiprocessing as mp
import os
import pickle
import socket
import sys
import traceback
from typing import Any, Callable, Optional


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker_entry(fn, kwargs, env, result_q):
    os.environ.update(env)
    rank = int(env["RANK"])
    try:
        from . import api

        api.init()
        try:
            result = fn(**kwargs)
        finally:
            api.shutdown()
        if rank == 0:
            # pre-pickle to plain bytes: torch's ForkingPickler shares tensor
            # storage by fd, which breaks once this worker exits
            result_q.put(("ok", pickle.dumps(result)))
    except BaseException:
        if rank == 0:
            result_q.put(("err", traceback.format_exc()))
        traceback.print_exc()
        sys.exit(1)


class Runner:
    """``Runner(np=8).run(train_fn, **kwargs)``.

    np = -1 : run fn() once in-process (world of 1, no process group);
    np >= 1 : spawn np local worker processes, return rank-0's result.
    """

    def __init__(
        self,
        np: int,
        driver_log_verbosity: str = "all",
        timeout_s: float = 3600.0,
        max_restarts: int = 0,
    ):
        self.np = np
        self.timeout_s = timeout_s
        # the reference delegates fault tolerance to Spark
        # barrier mode = gang-scheduled all-or-nothing job retry. Equivalent
        # here: on any worker failure the gang is aborted and, with
        # max_restarts > 0, relaunched whole; the train fn resumes from its
        # last checkpoint (ModelCheckpoint + the broadcast callback restore
        # rank parity, §5.4). Workers see DDLW_RESTART_ATTEMPT in env.
        self.max_restarts = max_restarts

    def run(self, fn: Callable[..., Any], **kwargs) -> Any:
        if self.np == -1:
            from . import api

            api.shutdown()  # ensure clean world-of-1 state
            return fn(**kwargs)
        if self.np < 1:
            raise ValueError(f"np must be -1 or >= 1, got {self.np}")
        last_exc: Optional[BaseException] = None
        for attempt in range(1 + self.max_restarts):
            try:
                return self._run_once(fn, kwargs, attempt)
            except (RuntimeError, TimeoutError) as e:
                last_exc = e
                if attempt < self.max_restarts:
                    print(
                        f"[ddlw.Runner] job attempt {attempt} failed ({e}); "
                        f"restarting gang ({self.max_restarts - attempt} retries left)",
                        file=sys.stderr,
                        flush=True,
                    )
        assert last_exc is not None
        raise last_exc

    def _run_once(self, fn: Callable[..., Any], kwargs: dict, attempt: int) -> Any:
        ctx = mp.get_context("spawn")
        result_q = ctx.SimpleQueue()
        port = _free_port()
        procs = []
        for r in range(self.np):
            env = {
                "RANK": str(r),
                "LOCAL_RANK": str(r),
                "WORLD_SIZE": str(self.np),
                "MASTER_ADDR": "127.0.0.1",
                "MASTER_PORT": str(port),
                "DDLW_RESTART_ATTEMPT": str(attempt),
            }
            p = ctx.Process(target=_worker_entry, args=(fn, kwargs, env, result_q), daemon=False)
            p.start()
            procs.append(p)

        result: Optional[tuple] = None
        try:
            # poll the gang: ANY worker death aborts promptly (a sequential
            # join would wait the earlier ranks' full timeout before
            # noticing a later rank died while rank 0 blocks in a
            # collective); one shared deadline for the whole gang
            import time as _time

            deadline = _time.monotonic() + self.timeout_s
            while True:
                for p in procs:
                    if p.exitcode not in (0, None):
                        raise RuntimeError(
                            f"worker (pid {p.pid}) exited with code "
                            f"{p.exitcode}; aborting job"
                        )
                if not any(p.is_alive() for p in procs):
                    break
                if _time.monotonic() > deadline:
                    raise TimeoutError(
                        f"workers still alive after {self.timeout_s}s; aborting")
                _time.sleep(0.2)
            if not result_q.empty():
                result = result_q.get()
        finally:
            for p in procs:
                if p.is_alive():
                    p.terminate()
            for p in procs:
                p.join(5)

        if result is None:
            raise RuntimeError("rank 0 produced no result")
        status, payload = result
        if status == "err":
            raise RuntimeError(f"rank 0 raised:\n{payload}")
        return pickle.loads(payload)
