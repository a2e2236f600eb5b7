"""Synthetic Europarl-shape corpus generator (device-resident).

BASELINE.md names the reference headline on Europarl v7 English: 49,158,635
running words, 1,965,734 lines, 197 splits (README.md:43-48).  There is no
network for the real corpus, so we synthesize text of the same shape:
Zipf-distributed word frequencies over a ~130k vocabulary with natural word
lengths, whitespace-separated, split on word boundaries into 197 splits.
Generation happens once per run on the GPU (not timed)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import List

import numpy as np
import torch

EUROPARL_WORDS = 49_158_635
EUROPARL_SPLITS = 197
EUROPARL_VOCAB = 130_000


@dataclass
class Corpus:
    text: torch.Tensor          # u8[nbytes], device-resident
    split_offsets: List[int]    # byte offset of each split start (+ end)
    nwords: int

    @property
    def nbytes(self) -> int:
        return int(self.text.numel())

    def splits(self):
        return list(zip(self.split_offsets[:-1], self.split_offsets[1:]))


def _build_vocab(vocab_size: int, seed: int):
    """Random lowercase words, lognormal-ish lengths (mean ~4.8 chars like
    English running text), deduplicated."""
    rng = np.random.default_rng(seed)
    lens = np.clip(rng.lognormal(mean=1.45, sigma=0.45,
                                 size=int(vocab_size * 1.3)).astype(np.int64),
                   1, 15)
    letters = rng.integers(ord("a"), ord("z") + 1,
                           size=int(lens.sum()), dtype=np.uint8)
    words = []
    seen = set()
    off = 0
    for L in lens:
        w = letters[off:off + L].tobytes()
        off += L
        if w not in seen:
            seen.add(w)
            words.append(w)
        if len(words) == vocab_size:
            break
    # ensure exactly vocab_size entries (append numbered words if collisions
    # exhausted the pool)
    i = 0
    while len(words) < vocab_size:
        w = b"w%d" % i
        if w not in seen:
            seen.add(w)
            words.append(w)
        i += 1
    return words


def make_corpus(device, nwords: int = EUROPARL_WORDS,
                nsplits: int = EUROPARL_SPLITS,
                vocab_size: int = EUROPARL_VOCAB, seed: int = 0,
                zipf_s: float = 1.07) -> Corpus:
    """Build the corpus directly on `device`.

    Word ids are sampled by inverse-CDF from a Zipf(s) distribution (the
    empirical shape of natural-language unigram frequencies); bytes are
    assembled with the gather_bytes HIP kernel on GPU (CPU fallback for
    test environments)."""
    from .. import ops

    words = _build_vocab(vocab_size, seed + 1)
    # vocab blob: each entry = word bytes + ' ' separator
    lens = np.array([len(w) + 1 for w in words], dtype=np.int64)
    blob = b"".join(w + b" " for w in words)
    voff = np.concatenate([[0], np.cumsum(lens)[:-1]])

    dev = torch.device(device)
    vocab_blob = torch.from_numpy(
        np.frombuffer(blob, dtype=np.uint8).copy()).to(dev)
    # zipf CDF over ranks 1..V
    ranks = np.arange(1, vocab_size + 1, dtype=np.float64)
    p = ranks ** (-zipf_s)
    cdf = torch.from_numpy(np.cumsum(p / p.sum())).to(dev)

    g = torch.Generator(device=dev)
    g.manual_seed(seed)
    # chunked sampling: ATen's searchsorted rejects >2^31-element launches
    ids = torch.empty(nwords, dtype=torch.int64, device=dev)
    CH = 1 << 29
    for off in range(0, nwords, CH):
        m = min(CH, nwords - off)
        u = torch.rand(m, generator=g, device=dev, dtype=torch.float64)
        torch.searchsorted(cdf, u, out=ids[off:off + m])
    ids.clamp_(max=vocab_size - 1)

    lens_t = torch.from_numpy(lens).to(dev)
    voff_t = torch.from_numpy(voff.astype(np.int64)).to(dev)
    wlens = lens_t.index_select(0, ids)
    wstarts = voff_t.index_select(0, ids)
    out_off = torch.cumsum(wlens, 0) - wlens
    total = int((out_off[-1] + wlens[-1]).item())
    # pack (start << 16 | len) for gather_bytes
    pos = (wstarts << 16) | wlens
    if dev.type == "cuda":
        text = ops.ext().gather_bytes(vocab_blob, pos, out_off, total)
    else:
        from ..ops import _cpu
        _, text = _cpu.extract_words(vocab_blob, pos)
    # split boundaries on word-count boundaries -> byte offsets
    word_bounds = [round(i * nwords / nsplits) for i in range(nsplits)]
    off_host = out_off.index_select(
        0, torch.tensor(word_bounds, device=dev, dtype=torch.int64)).cpu()
    split_offsets = [int(x) for x in off_host] + [total]
    return Corpus(text=text, split_offsets=split_offsets, nwords=nwords)
