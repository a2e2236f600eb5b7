"""Negative tests for the user-program contract checks:
duplicate taskfn keys (server.lua:258-261), oversized taskfn values
(server.lua:262-267, MAX_TASKFN_VALUE_SIZE), integer partitionfn results
(job.lua:203-206), and missing mandatory roles (server.lua:427-428)."""

import pytest

from mapreduce_amd import run_local
from mapreduce_amd.utils import MAX_TASKFN_VALUE_SIZE


def _fns(**over):
    base = {
        "taskfn": lambda emit: emit("k1", "a b"),
        "mapfn": lambda k, v, emit: [emit(w, 1) for w in v.split()],
        "partitionfn": lambda k: 0,
        "reducefn": lambda k, vs, emit: emit(sum(vs)),
        "associative_reducer": True,
        "commutative_reducer": True,
        "idempotent_reducer": True,
    }
    base.update(over)
    return {r: base for r in ("taskfn", "mapfn", "partitionfn", "reducefn")}


def test_missing_mandatory_role():
    fns = _fns()
    del fns["reducefn"]
    with pytest.raises(ValueError, match="reducefn"):
        run_local({"fns": fns, "verbose": False})


def test_duplicate_taskfn_key():
    def taskfn(emit):
        emit("same", "x")
        emit("same", "y")

    with pytest.raises(ValueError, match="duplicate taskfn key"):
        run_local({"fns": _fns(taskfn=taskfn), "verbose": False})


def test_oversized_taskfn_value():
    big = "z" * (MAX_TASKFN_VALUE_SIZE + 1)
    with pytest.raises(ValueError, match="exceeds"):
        run_local({"fns": _fns(taskfn=lambda emit: emit("k", big)),
                   "verbose": False})


def test_non_integer_partitionfn():
    fns = _fns(partitionfn=lambda k: "zero")
    # the worker's crash barrier marks the job BROKEN; after
    # MAX_JOB_RETRIES the server promotes it to FAILED and the task
    # still completes (server.lua:192-205 semantics)
    srv = run_local({"fns": fns, "verbose": False})
    assert srv.finished
    assert srv.stats["map_failed"] >= 1
