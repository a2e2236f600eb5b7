"""In-process A/B of MR_SPILL_CHUNK (tokenizer spill allocator chunk size).

One corpus, one process: the env var is read per kernel launch, so
flipping os.environ between timed blocks compares instantiations without
paying corpus generation per setting.
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
    for setting in sys.argv[1:] or ["512", "1024", "2048", "512"]:
        os.environ["MR_SPILL_CHUNK"] = setting
        for _ in range(warm):
            job.run(corpus.text, splits)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(steps):
            job.run(corpus.text, splits)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / steps * 1000
        # reserved spill entries (real misses + HT_EMPTY chunk-tail pads):
        # re-run the map phase alone to read the counter
        job.begin_map(corpus.text)
        job.map_split(splits[0][0], splits[-1][1])
        reserved = int(job._spill_c.sum().item())
        nreal = int(job._nwords.item())
        print(f"MR_SPILL_CHUNK={setting}: {ms:.3f} ms/step "
              f"(reserved {reserved/1e6:.2f}M spill entries, "
              f"{nreal/1e6:.1f}M words)", flush=True)


if __name__ == "__main__":
    main()
