"""Per-module unit tests: fs backends round-trip (fs.lua:213-251 utest
parity) and claim atomicity under contention (task.lua:301-309)."""

import threading

import pytest

import mapreduce_amd.fs as fsmod
from mapreduce_amd.parallel.coord import LocalCoordinator
from mapreduce_amd.task import Task, make_job
from mapreduce_amd.utils import STATUS, TASK_STATUS


@pytest.mark.parametrize("storage", ["mem:utest", "shared:{tmp}/fsround"])
def test_fs_roundtrip(storage, tmp_path):
    fs = fsmod.router(storage.format(tmp=tmp_path))
    rows = [("a", [1, 2]), ((1, "x"), ["v"]), (3.5, [None])]
    b = fs.builder("map_results.P0.M1")
    for k, v in rows:
        b.append(k, v)
    b.build()
    b2 = fs.builder("map_results.P1.M1")
    b2.append("z", [9])
    b2.build()
    assert fs.list(r"^map_results\.P0\..*$") == ["map_results.P0.M1"]
    assert sorted(fs.list(r"^map_results\..*$")) == [
        "map_results.P0.M1", "map_results.P1.M1"]
    assert list(fs.records("map_results.P0.M1")) == rows
    # idempotent republish (retry semantics, job.lua:219)
    b3 = fs.builder("map_results.P0.M1")
    b3.append("only", [1])
    b3.build()
    assert list(fs.records("map_results.P0.M1")) == [("only", [1])]
    fs.remove("map_results.P0.M1")
    assert fs.list(r"^map_results\.P0\..*$") == []
    fs.cleanup()


def test_claim_atomicity_under_contention():
    """N threads race to claim M jobs; every job is claimed exactly once."""
    coord = LocalCoordinator()
    task0 = Task(coord)
    task0.create_collection(TASK_STATUS.MAP, {
        "fns": {}, "storage": "mem:claim", "result_ns": "result"}, 1)
    M = 40
    task0.insert_jobs(Task.MAP_JOBS, [make_job(str(i), i) for i in range(M)])

    wins = []
    lock = threading.Lock()

    def claimer(name):
        t = Task(coord)
        t.update()
        while True:
            ns, doc = t.take_next_job(name, name)
            if doc is None:
                return
            with lock:
                wins.append((doc["_id"], name))

    threads = [threading.Thread(target=claimer, args=(f"w{i}",))
               for i in range(8)]
    for th in threads:
        th.start()
    for th in threads:
        th.join()
    ids = [w[0] for w in wins]
    assert sorted(ids) == sorted(str(i) for i in range(M))
    assert len(set(ids)) == M  # exactly-once claims
    docs = task0.scan_jobs(Task.MAP_JOBS)
    assert all(d["status"] == STATUS.RUNNING for d in docs)


def test_broken_reclaim_increments_nothing_on_claim():
    """Claiming a BROKEN job does not bump repetitions (only crashes do,
    job.lua:322-342)."""
    coord = LocalCoordinator()
    t = Task(coord)
    t.create_collection(TASK_STATUS.MAP, {
        "fns": {}, "storage": "mem:b", "result_ns": "result"}, 1)
    j = make_job("1", 1)
    j["status"] = STATUS.BROKEN
    j["repetitions"] = 1
    t.insert_jobs(Task.MAP_JOBS, [j])
    ns, doc = t.take_next_job("w", "w")
    assert doc["status"] == STATUS.RUNNING
    assert doc["repetitions"] == 1


def test_exhausted_broken_not_claimable():
    from mapreduce_amd.utils import MAX_JOB_RETRIES
    coord = LocalCoordinator()
    t = Task(coord)
    t.create_collection(TASK_STATUS.MAP, {
        "fns": {}, "storage": "mem:c", "result_ns": "result"}, 1)
    j = make_job("1", 1)
    j["status"] = STATUS.BROKEN
    j["repetitions"] = MAX_JOB_RETRIES
    t.insert_jobs(Task.MAP_JOBS, [j])
    ns, doc = t.take_next_job("w", "w")
    assert doc is None
    assert t.promote_broken(Task.MAP_JOBS) == 1
    d, _ = coord.get_doc(f"{Task.MAP_JOBS}/1")
    assert d["status"] == STATUS.FAILED
