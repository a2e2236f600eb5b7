"""TeraSort benchmark: 64-bit keys + 64-bit payloads (BASELINE.json
config 4: "TeraSort 10 GB synthetic").  Single-rank by default; under
torchrun the data is sharded per rank (weak scaling).

    python benchmarks/terasort_bench.py --gb 10 --steps 5 --warmup 2

Also A/Bs the hand-written radix sort against torch.sort (rocPRIM) on the
same data — torch sorts signed int64 so its ORDER differs, but the work is
comparable; reported for context.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gb", type=float, default=10.0,
                   help="total GB of (key,payload) pairs per rank")
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--torch-ab", action="store_true",
                   help="also time torch.sort on the same keys")
    args = p.parse_args()

    from mapreduce_amd.gpu import dist as dx
    from mapreduce_amd.gpu.terasort import TeraSortJob

    rank, world, device = dx.init_from_env()
    n = int(args.gb * (1 << 30) / 16)  # 16 B per (key, payload)
    g = torch.Generator(device=device)
    g.manual_seed(1234 + rank)
    keys = torch.randint(-2 ** 63, 2 ** 63 - 1, (n,), generator=g,
                         dtype=torch.int64, device=device)
    pay = torch.arange(n, dtype=torch.int64, device=device)
    job = TeraSortJob(device)

    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    for _ in range(args.warmup):
        sk, sv = job.run(keys, pay)
    sync()
    dx.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        sk, sv = job.run(keys, pay)
    sync()
    dx.barrier()
    el = time.perf_counter() - t0
    assert job.validate(sk)

    out = {
        "metric": "sorted pairs/sec (whole job)",
        "value": n * world * args.steps / el,
        "unit": "pairs/s",
        "n_gpus": world,
        "steps": args.steps,
        "ms_per_step": el / args.steps * 1000,
        "gb_per_rank": args.gb,
        "higher_is_better": True,
        "scaling": "weak",
        "dtype": "u64 keys + u64 payloads",
        "data": "synthetic uniform random",
    }
    if args.torch_ab and device.type == "cuda":
        for _ in range(2):
            torch.sort(keys)
        sync()
        t1 = time.perf_counter()
        for _ in range(args.steps):
            torch.sort(keys)  # keys only (signed order) — context number
        sync()
        out["torch_sort_keysonly_ms"] = (
            (time.perf_counter() - t1) / args.steps * 1000)
    if rank == 0:
        print(json.dumps(out), flush=True)


if __name__ == "__main__":
    sys.exit(main())
