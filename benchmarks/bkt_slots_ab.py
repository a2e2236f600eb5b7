"""In-process A/B of MR_BKT_SLOTS (bucket_count LDS table size).

PMC showed bucket_count latency-bound (waves wait ~16k cycles per ~660
VALU instrs) at 40 KB LDS = 4 blocks/CU; smaller tables raise occupancy.
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from mapreduce_amd.gpu.corpus import make_corpus  # noqa: E402
from mapreduce_amd.gpu.wordcount import WordCountJob  # noqa: E402


def main():
    dev = torch.device("cuda:0")
    corpus = make_corpus(dev, nwords=49_158_635, nsplits=197,
                         vocab_size=130_000, seed=1234)
    job = WordCountJob(dev, vocab_estimate=130_000, mode="streaming")
    splits = corpus.splits()
    steps, warm = 15, 5
    for setting in sys.argv[1:] or ["2048", "1024", "512", "2048"]:
        os.environ["MR_BKT_SLOTS"] = setting
        for _ in range(warm):
            job.run(corpus.text, splits)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(steps):
            job.run(corpus.text, splits)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / steps * 1000
        print(f"MR_BKT_SLOTS={setting}: {ms:.3f} ms/step", flush=True)


if __name__ == "__main__":
    main()
