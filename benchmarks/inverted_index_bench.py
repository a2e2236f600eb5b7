"""Inverted-index benchmark (BASELINE.json config 3: same Europarl-shape
corpus, large value lists stressing the all-to-all shuffle).

    python benchmarks/inverted_index_bench.py --steps 5 --warmup 2
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
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--words", type=int, default=49_158_635)
    p.add_argument("--splits", type=int, default=197,
                   help="documents per rank")
    p.add_argument("--vocab", type=int, default=130_000)
    args = p.parse_args()

    from mapreduce_amd.gpu import dist as dx
    from mapreduce_amd.gpu.corpus import make_corpus
    from mapreduce_amd.gpu.inverted_index import InvertedIndexJob

    rank, world, device = dx.init_from_env()
    corpus = make_corpus(device, nwords=args.words, nsplits=args.splits,
                         vocab_size=args.vocab, seed=77 + rank)
    job = InvertedIndexJob(device, doc_base=rank * args.splits)

    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    res = None
    for _ in range(args.warmup):
        res = job.run(corpus.text, corpus.splits())
    sync()
    dx.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        res = job.run(corpus.text, corpus.splits())
    sync()
    dx.barrier()
    el = time.perf_counter() - t0
    npostings = int(res.tf.numel())
    out = {
        "metric": "indexed words/sec (whole job)",
        "value": args.words * world * args.steps / el,
        "unit": "words/s",
        "n_gpus": world,
        "steps": args.steps,
        "ms_per_step": el / args.steps * 1000,
        "postings_per_rank": npostings,
        "unique_words_per_rank": int(res.keys.numel()),
        "higher_is_better": True,
        "scaling": "weak",
        "data": f"synthetic Europarl shape ({args.words} words, "
                f"{args.splits} docs/rank, Zipf vocab {args.vocab})",
    }
    if rank == 0:
        print(json.dumps(out), flush=True)


if __name__ == "__main__":
    sys.exit(main())
