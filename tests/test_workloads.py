"""TeraSort + inverted index on the CPU tier (engine logic; GPU numerics
covered by test_gpu_workloads.py), single-rank and gloo world_size=2."""

import collections
import os
import socket

import numpy as np
import pytest
import torch

from mapreduce_amd.gpu.inverted_index import InvertedIndexJob
from mapreduce_amd.gpu.terasort import TeraSortJob


def _u64(t):
    return t.numpy().view(np.uint64)


def test_terasort_single_rank_cpu():
    rng = np.random.default_rng(5)
    keys = rng.integers(0, 2 ** 64 - 1, size=50_000, dtype=np.uint64)
    pay = np.arange(50_000, dtype=np.uint64)
    job = TeraSortJob("cpu")
    sk, sv = job.run(torch.from_numpy(keys.view(np.int64)),
                     torch.from_numpy(pay.view(np.int64)))
    order = np.argsort(keys, kind="stable")
    assert np.array_equal(_u64(sk), keys[order])
    assert np.array_equal(_u64(sv), pay[order])
    assert job.validate(sk)


def py_inverted_index(docs):
    idx = {}
    for d, text in enumerate(docs):
        for w in text.split():
            ent = idx.setdefault(w, {})
            ent[d] = ent.get(d, 0) + 1
    return {w: sorted(v.items()) for w, v in idx.items()}


def test_inverted_index_single_rank_cpu():
    rng = np.random.default_rng(9)
    vocab = [f"tok{i}".encode() for i in range(200)]
    docs = []
    for _ in range(6):
        ids = rng.integers(0, len(vocab), size=500)
        docs.append(b" ".join(vocab[i] for i in ids.tolist()))
    blob = b" ".join(docs) + b" "
    # split offsets: each doc starts right after the previous separator
    offs = [0]
    for d in docs[:-1]:
        offs.append(offs[-1] + len(d) + 1)
    offs.append(len(blob))
    splits = list(zip(offs[:-1], offs[1:]))
    text = torch.from_numpy(np.frombuffer(blob, dtype=np.uint8).copy())
    job = InvertedIndexJob("cpu")
    res = job.run(text, splits)
    got = res.to_host()
    exp = py_inverted_index(docs)
    assert got == exp
    # lookup() serves single-word postings off the sorted index
    for w in (vocab[0], vocab[57], vocab[199]):
        assert res.lookup(w) == exp[w]
    assert res.lookup("absent-word") == []


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _ts_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    try:
        rng = np.random.default_rng(40 + rank)
        keys = rng.integers(0, 2 ** 64 - 1, size=20_000, dtype=np.uint64)
        job = TeraSortJob("cpu")
        sk, _ = job.run(torch.from_numpy(keys.view(np.int64)), None)
        assert job.validate(sk)
        mine = _u64(sk)
        # ownership: my keys' top bytes map to my rank
        if len(mine):
            tb = (mine >> np.uint64(56)).astype(np.int64)
            assert ((tb * world) >> 8 == rank).all()
        # global preservation
        all_keys = [None] * world
        torch.distributed.all_gather_object(all_keys, mine.tolist())
        all_in = [None] * world
        torch.distributed.all_gather_object(all_in, keys.tolist())
        if rank == 0:
            got = sorted(x for l in all_keys for x in l)
            exp = sorted(x for l in all_in for x in l)
            assert got == exp
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(180)
def test_terasort_gloo_ws2():
    torch.multiprocessing.spawn(_ts_worker, args=(2, _free_port()),
                                nprocs=2, join=True)


def _ii_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    try:
        rng = np.random.default_rng(70 + rank)
        vocab = [f"w{i}".encode() for i in range(60)]
        docs = []
        for _ in range(3):
            ids = rng.integers(0, len(vocab), size=200)
            docs.append(b" ".join(vocab[i] for i in ids.tolist()))
        blob = b" ".join(docs) + b" "
        offs = [0]
        for d in docs[:-1]:
            offs.append(offs[-1] + len(d) + 1)
        offs.append(len(blob))
        splits = list(zip(offs[:-1], offs[1:]))
        text = torch.from_numpy(np.frombuffer(blob, dtype=np.uint8).copy())
        job = InvertedIndexJob("cpu", doc_base=rank * 3)
        res = job.run(text, splits)
        part = res.to_host()
        alldocs = [None] * world
        torch.distributed.all_gather_object(alldocs, docs)
        allparts = [None] * world
        torch.distributed.all_gather_object(allparts, part)
        if rank == 0:
            merged = {}
            for p in allparts:
                for w, lst in p.items():
                    assert w not in merged, "word owned by two ranks"
                    merged[w] = lst
            flat = [d for dl in alldocs for d in dl]
            assert merged == py_inverted_index(flat)
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(180)
def test_inverted_index_gloo_ws2():
    torch.multiprocessing.spawn(_ii_worker, args=(2, _free_port()),
                                nprocs=2, join=True)


def _ts_sample_worker(rank, world, port):
    """Skewed keys (all top bytes 0) — "topbyte" would send everything to
    rank 0; sampled splitters balance the ranks and preserve global
    order (rank-major by splitter range)."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    try:
        rng = np.random.default_rng(90 + rank)
        n = 20_000
        keys = rng.integers(0, 2 ** 40, size=n, dtype=np.uint64)  # top byte 0
        pl = rng.integers(0, 2 ** 62, size=n, dtype=np.uint64)
        job = TeraSortJob("cpu", partitioner="sample")
        sk, sv = job.run(torch.from_numpy(keys.view(np.int64)),
                         torch.from_numpy(pl.view(np.int64)))
        assert job.validate(sk)
        mine = _u64(sk)
        # balance: no rank holds more than 65% of the total
        sizes = [None] * world
        torch.distributed.all_gather_object(sizes, len(mine))
        assert max(sizes) <= 0.65 * sum(sizes), sizes
        # payloads still ride with their keys (post-exchange a rank holds
        # keys from EVERY rank — pair against the gathered global map)
        pairs = [None] * world
        torch.distributed.all_gather_object(
            pairs, list(zip(keys.tolist(), pl.tolist())))
        kp = {k: v for plist in pairs for k, v in plist}
        pv = _u64(sv)
        for i in range(0, len(mine), max(1, len(mine) // 64)):
            assert kp[int(mine[i])] == int(pv[i])
        # global order: every key on rank r sorts <= every key on rank r+1
        ends = [None] * world
        torch.distributed.all_gather_object(
            ends, (int(mine[0]), int(mine[-1])) if len(mine) else None)
        all_keys = [None] * world
        torch.distributed.all_gather_object(all_keys, mine.tolist())
        all_in = [None] * world
        torch.distributed.all_gather_object(all_in, keys.tolist())
        if rank == 0:
            prev_end = None
            for e in ends:
                if e is None:
                    continue
                if prev_end is not None:
                    assert prev_end <= e[0]
                prev_end = e[1]
            got = sorted(x for l in all_keys for x in l)
            exp = sorted(x for l in all_in for x in l)
            assert got == exp
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(180)
def test_terasort_sampled_splitters_gloo_ws2():
    torch.multiprocessing.spawn(_ts_sample_worker, args=(2, _free_port()),
                                nprocs=2, join=True)


def test_terasort_sample_single_rank():
    rng = np.random.default_rng(4)
    keys = rng.integers(0, 2 ** 30, size=5_000, dtype=np.uint64)
    job = TeraSortJob("cpu", partitioner="sample")
    sk, _ = job.run(torch.from_numpy(keys.view(np.int64)), None)
    assert job.validate(sk)
    assert np.array_equal(_u64(sk), np.sort(keys))


def test_extremes_example():
    """Per-key min/max task script (idempotent reducer) vs a direct
    oracle; the same module serves every role (INIT-SCRIPT case)."""
    from mapreduce_amd import run_local
    from mapreduce_amd.examples import extremes

    extremes.init({"nstations": 6, "n": 100, "seed": 3})
    srv = run_local({"fns": {r: extremes for r in (
        "taskfn", "mapfn", "partitionfn", "reducefn", "combinerfn",
        "finalfn")}, "verbose": False}, nworkers=2)
    assert srv.finished
    exp = {s: (min(v), max(v)) for s, v in extremes.CONF["readings"].items()}
    assert extremes.RESULTS == exp


def _keyed_oracle(keys, vals, op):
    import numpy as np
    agg = {}
    f = {"sum": lambda a, b: a + b, "min": min, "max": max}[op]
    for k, v in zip(keys.tolist(), vals.tolist()):
        agg[k] = f(agg[k], v) if k in agg else v
    uk = np.sort(np.array(list(agg), dtype=np.uint64))
    return uk, np.array([agg[int(k)] for k in uk])


def test_keyed_reduce_single_rank():
    from mapreduce_amd.gpu.keyed_reduce import KeyedReduceJob

    rng = np.random.default_rng(8)
    keys = rng.integers(0, 500, size=30_000, dtype=np.uint64) * 7919
    for op in ("sum", "min", "max"):
        for vals_np in (rng.integers(-10 ** 9, 10 ** 9, size=30_000,
                                     dtype=np.int64),
                        rng.standard_normal(30_000)):
            job = KeyedReduceJob("cpu", op=op)
            uk, uv = job.run(torch.from_numpy(keys.view(np.int64)),
                             torch.from_numpy(vals_np))
            ek, ev = _keyed_oracle(keys, vals_np, op)
            assert np.array_equal(_u64(uk), ek)
            if vals_np.dtype == np.int64:
                assert np.array_equal(uv.numpy(), ev)
            else:
                assert np.allclose(uv.numpy(), ev, rtol=1e-12, atol=1e-9)


def _kr_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from mapreduce_amd.gpu.keyed_reduce import KeyedReduceJob

        rng = np.random.default_rng(60 + rank)
        keys = rng.integers(0, 300, size=10_000, dtype=np.uint64) * 104729
        vals = rng.integers(-10 ** 6, 10 ** 6, size=10_000, dtype=np.int64)
        job = KeyedReduceJob("cpu", op="min")
        uk, uv = job.run(torch.from_numpy(keys.view(np.int64)),
                         torch.from_numpy(vals))
        # ownership: mulhi partition
        parts = ((_u64(uk).astype(object) * world) >> 64).astype(int)
        assert (parts == rank).all()
        all_pairs = [None] * world
        torch.distributed.all_gather_object(
            all_pairs, (_u64(uk).tolist(), uv.tolist()))
        all_in = [None] * world
        torch.distributed.all_gather_object(
            all_in, (keys.tolist(), vals.tolist()))
        if rank == 0:
            got = {}
            for ks, vs in all_pairs:
                for k, v in zip(ks, vs):
                    assert k not in got, "key owned by two ranks"
                    got[k] = v
            exp = {}
            for ks, vs in all_in:
                for k, v in zip(ks, vs):
                    exp[k] = min(exp.get(k, v), v)
            assert got == exp
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(180)
def test_keyed_reduce_gloo_ws2():
    torch.multiprocessing.spawn(_kr_worker, args=(2, _free_port()),
                                nprocs=2, join=True)
