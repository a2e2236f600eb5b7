"""Control-plane-driven engine runs (gpu.runner) on the CPU tier."""

import collections
import os
import socket

import pytest
import torch

from mapreduce_amd.gpu.corpus import make_corpus
from mapreduce_amd.gpu.runner import GpuClusterRunner
from mapreduce_amd.gpu.wordcount import WordCountJob
from mapreduce_amd.utils import STATUS


def oracle(text_bytes):
    return dict(collections.Counter(text_bytes.split()))


@pytest.mark.parametrize("claim_mode", ["batch", "dynamic"])
def test_runner_single_rank(claim_mode):
    c = make_corpus("cpu", nwords=6000, nsplits=5, vocab_size=300, seed=2)
    job = WordCountJob("cpu", vocab_estimate=600)
    runner = GpuClusterRunner(job, claim_mode=claim_mode)
    res = runner.run(c.text, c.splits())
    assert res.nwords == 6000
    assert dict(res.to_host()) == oracle(bytes(c.text.numpy().tobytes()))
    st = runner.job_stats()
    if claim_mode == "dynamic":
        assert st == {"jobs": 5, "written": 5, "broken": 0,
                      "shuffle_rounds": 1}
    else:
        assert st["status"] == STATUS.WRITTEN


def test_runner_phase_retry_after_transient_fault():
    """A transient map-split failure triggers a PHASE-scoped retry (device
    state reset + exact replay) and still yields correct counts."""
    c = make_corpus("cpu", nwords=4000, nsplits=4, vocab_size=200, seed=9)

    class FlakyJob(WordCountJob):
        def __init__(self, *a, **k):
            super().__init__(*a, **k)
            self.calls = 0

        def map_split(self, s, e):
            self.calls += 1
            if self.calls == 2:
                raise RuntimeError("injected transient device fault")
            super().map_split(s, e)

    job = FlakyJob("cpu", vocab_estimate=400)
    runner = GpuClusterRunner(job, claim_mode="dynamic")
    res = runner.run(c.text, c.splits())
    assert res.nwords == 4000
    assert dict(res.to_host()) == oracle(bytes(c.text.numpy().tobytes()))
    assert job.calls >= 5  # first attempt partial + full replay


def _worker(rank, world, port, claim_mode):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    try:
        c = make_corpus("cpu", nwords=5000, nsplits=4, vocab_size=250,
                        seed=300 + rank)
        job = WordCountJob("cpu", vocab_estimate=600)
        runner = GpuClusterRunner(job, claim_mode=claim_mode)
        res = runner.run(c.text, c.splits())
        pairs = dict(res.to_host())
        allp = [None] * world
        torch.distributed.all_gather_object(allp, pairs)
        allt = [None] * world
        torch.distributed.all_gather_object(
            allt, bytes(c.text.numpy().tobytes()))
        if rank == 0:
            merged = {}
            for p in allp:
                for w, n in p.items():
                    assert w not in merged
                    merged[w] = n
            exp = collections.Counter()
            for t in allt:
                exp.update(t.split())
            assert merged == dict(exp)
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(180)
@pytest.mark.parametrize("claim_mode", ["batch", "dynamic"])
def test_runner_gloo_ws2(claim_mode):
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    torch.multiprocessing.spawn(_worker, args=(2, port, claim_mode),
                                nprocs=2, join=True)
