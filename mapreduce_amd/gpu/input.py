"""K8: host -> HBM input streaming.

The reference streams splits from GridFS/disk line-by-line
(utils.lua:133-200).  The MI355X equivalent: read files into PINNED host
staging buffers and overlap H2D copies on a side stream while the previous
chunk is being tokenized — input upload never serializes against compute.

load_corpus(): one-shot load of a file set into a device-resident corpus
(whitespace-aligned split boundaries preserved per file).
StreamLoader: double-buffered chunk iterator for corpora larger than the
wanted HBM footprint (map jobs consume chunk c while chunk c+1 uploads).
"""

from __future__ import annotations

import os
from typing import Iterator, List, Optional, Tuple

import torch

from .corpus import Corpus


def load_corpus(paths: List[str], device) -> Corpus:
    """Read files into one device corpus; each file = one split (the
    reference's taskfn-emits-files contract, examples/WordCount)."""
    dev = torch.device(device)
    sizes = [os.path.getsize(p) for p in paths]
    total = sum(sizes) + len(paths)  # +1 separator byte per file
    pinned = torch.empty(total, dtype=torch.uint8,
                         pin_memory=(dev.type == "cuda"))
    view = pinned.numpy()
    offsets = [0]
    off = 0
    for p, sz in zip(paths, sizes):
        with open(p, "rb") as fh:
            data = fh.read()
        view[off:off + sz] = memoryview(data)
        off += sz
        view[off] = 0x20  # separator keeps split boundaries ws-aligned
        off += 1
        offsets.append(off)
    text = pinned.to(dev, non_blocking=True)
    if dev.type == "cuda":
        torch.cuda.current_stream(dev).synchronize()
    nwords = None  # unknown until tokenized
    return Corpus(text=text, split_offsets=offsets, nwords=nwords)


class StreamLoader:
    """Double-buffered file -> HBM chunk stream.

    Iterates (device_chunk, base_offset); chunk c+1's H2D copy runs on a
    side stream while the caller processes chunk c.  Chunks split on
    whitespace so tokenization across chunk boundaries stays exact."""

    def __init__(self, path: str, device, chunk_bytes: int = 64 << 20):
        self.path = path
        self.device = torch.device(device)
        self.chunk_bytes = chunk_bytes
        self._copy_stream = (torch.cuda.Stream(self.device)
                             if self.device.type == "cuda" else None)

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, int]]:
        """Zero-copy-ish pipeline: file.readinto(pinned ring buffer) ->
        async H2D on the side stream; the consumer's stream waits on the
        copy event, and a buffer is reused only after its previous copy
        completed."""
        dev = self.device
        use_cuda = dev.type == "cuda"
        size = os.path.getsize(self.path)
        bufs = [torch.empty(self.chunk_bytes, dtype=torch.uint8,
                            pin_memory=use_cuda) for _ in range(2)]
        buf_ev: list = [None, None]
        pending: Optional[Tuple[torch.Tensor, int,
                                Optional["torch.cuda.Event"]]] = None
        with open(self.path, "rb") as fh:
            base = 0
            carry = b""
            bi = 0
            while base < size or carry:
                if buf_ev[bi] is not None:
                    buf_ev[bi].synchronize()  # previous copy out of buffer
                    buf_ev[bi] = None
                buf = bufs[bi]
                mv = memoryview(buf.numpy())
                nc = len(carry)
                mv[:nc] = carry
                nread = fh.readinto(mv[nc:])
                total = nc + nread
                if total == 0:
                    break
                at_eof = (base + total) >= size
                if not at_eof:
                    # cut at the last whitespace so no word spans chunks:
                    # scan a 64 KB tail first, the whole chunk if the tail
                    # is one ws-free run (only a word longer than the
                    # chunk itself is unsplittable)
                    tail0 = max(0, total - 65536)
                    tail = bytes(mv[tail0:total])
                    cut = max(tail.rfind(b" "), tail.rfind(b"\n"),
                              tail.rfind(b"\t"))
                    if cut >= 0:
                        cut = tail0 + cut + 1
                    else:
                        whole = bytes(mv[:total])
                        cut = max(whole.rfind(b" "), whole.rfind(b"\n"),
                                  whole.rfind(b"\t"))
                        cut = total if cut < 0 else cut + 1
                else:
                    cut = total
                carry = bytes(mv[cut:total])
                if use_cuda:
                    with torch.cuda.stream(self._copy_stream):
                        d = buf[:cut].to(dev, non_blocking=True)
                        ev = torch.cuda.Event()
                        ev.record(self._copy_stream)
                    buf_ev[bi] = ev
                else:
                    d = buf[:cut].clone()
                    ev = None
                if pending is not None:
                    pd, pb, pev = pending
                    if pev is not None:
                        torch.cuda.current_stream(dev).wait_event(pev)
                    yield pd, pb
                pending = (d, base, ev)
                base += cut
                bi ^= 1
        if pending is not None:
            pd, pb, pev = pending
            if pev is not None:
                torch.cuda.current_stream(dev).wait_event(pev)
            yield pd, pb
