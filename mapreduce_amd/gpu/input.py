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

    def _read_chunks(self) -> Iterator[Tuple[bytes, int]]:
        size = os.path.getsize(self.path)
        with open(self.path, "rb") as fh:
            base = 0
            carry = b""
            while base + len(carry) < size or carry:
                want = self.chunk_bytes - len(carry)
                data = carry + fh.read(want)
                if not data:
                    return
                if base + len(data) < size:
                    # cut at the last whitespace so no word spans chunks
                    cut = max(data.rfind(b" "), data.rfind(b"\n"),
                              data.rfind(b"\t"))
                    if cut <= 0:
                        cut = len(data)  # one giant word: hand it over whole
                    else:
                        cut += 1
                else:
                    cut = len(data)
                yield data[:cut], base
                carry = data[cut:]
                base += cut
                if base >= size and not carry:
                    return

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, int]]:
        dev = self.device
        use_cuda = dev.type == "cuda"
        pending: Optional[Tuple[torch.Tensor, int, Optional[torch.cuda.Event]]] = None
        for raw, base in self._read_chunks():
            host = torch.frombuffer(bytearray(raw), dtype=torch.uint8)
            if use_cuda:
                host = host.pin_memory()
                with torch.cuda.stream(self._copy_stream):
                    d = host.to(dev, non_blocking=True)
                    ev = torch.cuda.Event()
                    ev.record(self._copy_stream)
            else:
                d = host
                ev = None
            if pending is not None:
                pd, pb, pev = pending
                if pev is not None:
                    torch.cuda.current_stream(dev).wait_event(pev)
                yield pd, pb
            pending = (d, base, ev)
        if pending is not None:
            pd, pb, pev = pending
            if pev is not None:
                torch.cuda.current_stream(dev).wait_event(pev)
            yield pd, pb
