"""End-to-end wordcount vs the naive oracle (test.sh:8-87 analogue).

The reference diffs multi-worker server output against misc/naive.lua over
the same inputs, across a config matrix: (1) combiner + flagged reducer,
(2) no combiner + flagged reducer, (3) no combiner + general reducer,
(4) single INIT-SCRIPT module."""

import collections
import os

import pytest

import mapreduce_amd.examples.wordcount as wc
from mapreduce_amd import run_local

TEXT = """the quick brown fox jumps over the lazy dog
pack my box with five dozen liquor jugs
how vexingly quick daft zebras jump
the five boxing wizards jump quickly
sphinx of black quartz judge my vow
the the the quick quick fox
"""


def naive_oracle(files):
    """misc/naive.lua:1-7: single-process wordcount."""
    vocab = collections.Counter()
    for f in files:
        with open(f) as fh:
            for line in fh:
                vocab.update(line.split())
    return dict(vocab)


@pytest.fixture()
def corpus(tmp_path):
    files = []
    for i in range(3):
        p = tmp_path / f"in{i}.txt"
        p.write_text(TEXT * (i + 1) + f"unique{i}\n")
        files.append(str(p))
    return files


def _run(files, fns, nworkers=3):
    wc.init({"files": files, "out": None})
    srv = run_local({"fns": fns}, nworkers=nworkers)
    assert srv.finished
    return dict(wc.RESULTS)


def test_init_script_all_roles(corpus):
    # config (4): one module provides all roles (test.sh:64-77)
    got = _run(corpus, {r: wc for r in
                        ("taskfn", "mapfn", "partitionfn", "reducefn",
                         "combinerfn", "finalfn")})
    assert got == naive_oracle(corpus)


def test_no_combiner_flagged_reducer(corpus):
    got = _run(corpus, {"taskfn": wc, "mapfn": wc, "partitionfn": wc,
                        "reducefn": wc, "finalfn": wc})
    assert got == naive_oracle(corpus)


def test_no_combiner_general_reducer(corpus):
    # config (3): reducer without property flags -> always-reduce path
    # (reducefn2 in the reference, job.lua:276-284)
    general = {
        "init": lambda arg: None,
        "reducefn": lambda key, values, emit: emit(sum(values)),
    }
    got = _run(corpus, {"taskfn": wc, "mapfn": wc, "partitionfn": wc,
                        "reducefn": general, "finalfn": wc})
    assert got == naive_oracle(corpus)


def test_combiner_with_shared_storage(corpus, tmp_path):
    wc.init({"files": corpus, "out": None})
    srv = run_local(
        {"fns": {r: wc for r in ("taskfn", "mapfn", "partitionfn",
                                 "reducefn", "combinerfn", "finalfn")},
         "storage": f"shared:{tmp_path}/shuffle"},
        nworkers=2)
    assert srv.finished
    assert dict(wc.RESULTS) == naive_oracle(corpus)
    # results + spills cleaned up (finalfn returned True)
    left = [n for n in os.listdir(tmp_path / "shuffle")]
    assert not [n for n in left if n.startswith("result")]


def test_single_worker(corpus):
    got = _run(corpus, {r: wc for r in
                        ("taskfn", "mapfn", "partitionfn", "reducefn",
                         "combinerfn", "finalfn")}, nworkers=1)
    assert got == naive_oracle(corpus)


def test_stats_populated(corpus):
    wc.init({"files": corpus, "out": None})
    srv = run_local({"fns": {r: wc for r in
                             ("taskfn", "mapfn", "partitionfn", "reducefn",
                              "combinerfn", "finalfn")}}, nworkers=2)
    assert srv.stats["map"]["jobs"] == len(corpus)
    assert srv.stats["reduce"]["jobs"] >= 1
    assert srv.stats["map_failed"] == 0
    assert srv.stats["total_time"] > 0
