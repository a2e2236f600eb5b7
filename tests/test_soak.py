"""Concurrency soak: many jobs, several workers, random transient crashes,
iterative loop — the claims/retry/affinity machinery under churn."""

import random
import threading

import pytest

from mapreduce_amd import run_local


@pytest.mark.timeout(240)
def test_soak_iterative_with_random_transient_faults():
    rng = random.Random(1234)
    attempts = {}
    lock = threading.Lock()
    state = {"iter": 0, "sums": []}
    NJOBS = 40
    ITERS = 3

    def taskfn(emit):
        for i in range(NJOBS):
            emit(i, i + 1)

    crashes = {"n": 0}

    def mapfn(key, value, emit):
        with lock:
            attempts[key] = attempts.get(key, 0) + 1
            # a bounded number of transient faults (workers give up after
            # MAX_WORKER_RETRIES DISTINCT failed jobs — worker.lua:133-137
            # — so unbounded fault injection would exhaust the pool)
            crash = (crashes["n"] < 8 and attempts[key] % 2 == 1
                     and rng.random() < 0.15)
            if crash:
                crashes["n"] += 1
        if crash:
            raise RuntimeError(f"transient fault on {key}")
        emit("sum", value)
        emit(("pair", int(key) % 4), 1)

    def reducefn(key, values, emit):
        emit(sum(values))

    def finalfn(pairs):
        got = {}
        for k, v in pairs:
            got[k if not hasattr(k, "_items") else tuple(k)] = v[0]
        state["sums"].append(got["sum"])
        state["iter"] += 1
        return "loop" if state["iter"] < ITERS else True

    fns = {
        "init": lambda a: None,
        "taskfn": taskfn, "mapfn": mapfn,
        "partitionfn": lambda k: hash(k) % 5,
        "reducefn": reducefn, "finalfn": finalfn,
        "associative_reducer": True, "commutative_reducer": True,
        "idempotent_reducer": True,
    }
    srv = run_local({"fns": {r: fns for r in (
        "taskfn", "mapfn", "partitionfn", "reducefn", "finalfn")},
        "verbose": False}, nworkers=4)
    assert srv.finished
    expected = sum(range(1, NJOBS + 1))
    assert state["sums"] == [expected] * ITERS
    assert srv.iteration == ITERS
    assert srv.stats["map_failed"] == 0  # every fault retried to success
