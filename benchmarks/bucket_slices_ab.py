"""In-process A/B of MR_BUCKET_SLICES under the current spill allocator.

Each bucket_count block LDS-counts one slice of one bucket and flushes its
distinct keys to the global table; more slices = better load balance but
~slices-fold redundant global flushes per bucket.  The original sweep
(8=3.19, 16=2.58, 32=2.38 ms/step) ran when chunk-tail pads skewed bucket
255 by ~2M entries; the 2048-entry allocator changed that.
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
    for setting in sys.argv[1:] or ["32", "16", "8", "64", "32"]:
        os.environ["MR_BUCKET_SLICES"] = setting
        for _ in range(warm):
            job.run(corpus.text, splits)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(steps):
            job.run(corpus.text, splits)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / steps * 1000
        print(f"MR_BUCKET_SLICES={setting}: {ms:.3f} ms/step", flush=True)


if __name__ == "__main__":
    main()
