"""Multi-process integration: separate worker processes over the TCPStore
control plane + shared-dir shuffle — the test.sh topology (server foreground,
workers detached) with real process boundaries."""

import collections
import os
import socket
import subprocess
import sys
import time

import pytest

import mapreduce_amd.examples.wordcount as wc
from mapreduce_amd.server import Server

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def naive_oracle(files):
    vocab = collections.Counter()
    for f in files:
        with open(f) as fh:
            for line in fh:
                vocab.update(line.split())
    return dict(vocab)


@pytest.mark.timeout(120)
def test_two_worker_processes(tmp_path):
    files = []
    for i in range(4):
        p = tmp_path / f"in{i}.txt"
        p.write_text(("alpha beta gamma delta epsilon\n"
                      "beta beta gamma zeta\n") * (20 * (i + 1))
                     + f"only{i}\n")
        files.append(str(p))

    port = free_port()
    cnn = f"tcp://127.0.0.1:{port}"
    storage = f"shared:{tmp_path}/shuffle"
    init_args = {"files": files, "out": None}

    env = dict(os.environ, PYTHONPATH=REPO)
    workers = [
        subprocess.Popen(
            [sys.executable, "-m", "mapreduce_amd.execute_worker", cnn,
             "wc", "--max-iter", "1000000", "--max-tasks", "1000000"],
            env=env, cwd=REPO)
        for _ in range(2)
    ]
    try:
        wcmod = "mapreduce_amd.examples.wordcount"
        srv = Server(cnn, "wc").configure({
            "fns": {r: wcmod for r in
                    ("taskfn", "mapfn", "partitionfn", "reducefn",
                     "combinerfn", "finalfn")},
            "storage": storage,
            "init_args": init_args,
        })
        wc.init(init_args)  # server-side finalfn uses the module config
        srv.loop()
        assert srv.finished
        assert dict(wc.RESULTS) == naive_oracle(files)
        assert srv.stats["map"]["jobs"] == 4
        # both workers participated? (not guaranteed, but at least one did
        # and all jobs are accounted)
        assert srv.stats["map_failed"] == 0
        assert srv.stats["reduce_failed"] == 0
    finally:
        for w in workers:
            w.terminate()
        for w in workers:
            try:
                w.wait(timeout=10)
            except subprocess.TimeoutExpired:
                w.kill()


@pytest.mark.timeout(120)
def test_worker_joins_late(tmp_path):
    """Elasticity: a worker that connects after the task started still picks
    up jobs (workers join/leave at any time, README.md:13-16)."""
    files = []
    for i in range(3):
        p = tmp_path / f"in{i}.txt"
        p.write_text("x y z x y x\n" * 50)
        files.append(str(p))
    port = free_port()
    cnn = f"tcp://127.0.0.1:{port}"
    init_args = {"files": files, "out": None}
    env = dict(os.environ, PYTHONPATH=REPO)
    wcmod = "mapreduce_amd.examples.wordcount"
    srv = Server(cnn, "wc2").configure({
        "fns": {r: wcmod for r in
                ("taskfn", "mapfn", "partitionfn", "reducefn", "finalfn")},
        "storage": f"shared:{tmp_path}/shuffle2",
        "init_args": init_args,
    })
    wc.init(init_args)

    import threading
    procs = []

    def launch_late():
        time.sleep(0.5)
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "mapreduce_amd.execute_worker", cnn,
             "wc2", "--max-iter", "1000000", "--max-tasks", "1000000"],
            env=env, cwd=REPO))

    t = threading.Thread(target=launch_late)
    t.start()
    try:
        srv.loop()
        t.join()
        assert dict(wc.RESULTS) == naive_oracle(files)
    finally:
        t.join()
        for w in procs:
            w.terminate()
            try:
                w.wait(timeout=10)
            except subprocess.TimeoutExpired:
                w.kill()
