"""Intermediate (shuffle) storage — pluggable router, host tier.

Parity with mapreduce/fs.lua (router :185-208): a uniform
list/remove/builder/record-iterator facade over backends.  The reference's
backends are gridfs (Mongo), shared (NFS dir) and sshfs (scp pull); all are
file mediated.  Here:

  * "mem"            in-process dict of record blobs (single-process runs,
                     unit tests — fastest host tier)
  * "shared[:path]"  a directory on a (shared) filesystem — the
                     multi-process host tier, equivalent to fs.lua sharedfs
                     :119-137 (tmpfile + atomic rename like file_builder
                     :80-115)
  * the GPU tier does NOT live here: map outputs stay HBM-resident and move
    via RCCL all-to-all (mapreduce_amd.gpu.dist.exchange), replacing C5/C6 of
    SURVEY.md §2.5.

Files are streams of length-prefixed pickled (key, values) records sorted by
utils.sort_key — the binary replacement of the reference's "return k,{v}\n"
text rows (K7).
"""

from __future__ import annotations

import io
import os
import re
import tempfile
import threading
from typing import Any, Dict, Iterator, List, Tuple

from .utils import read_records, write_record

# registry for the "mem" backend: path -> {name -> bytes}
_MEM: Dict[str, Dict[str, bytes]] = {}
_MEM_LOCK = threading.Lock()


class Builder:
    """Buffered record writer with atomic publish (fs.lua file_builder
    :80-115: write tmpfile, rename on build — re-execution of a retried job
    simply republishes, keeping output idempotent per job.lua:219)."""

    def __init__(self, fs: "FS", name: str):
        self._fs = fs
        self._name = name
        self._buf = io.BytesIO()

    def append(self, key: Any, values: list) -> None:
        write_record(self._buf, key, values)

    def build(self) -> None:
        self._fs._publish(self._name, self._buf.getvalue())
        self._buf = io.BytesIO()


class FS:
    def builder(self, name: str) -> Builder:
        return Builder(self, name)

    def list(self, pattern: str) -> List[str]:
        raise NotImplementedError

    def remove(self, name: str) -> None:
        raise NotImplementedError

    def records(self, name: str) -> Iterator[Tuple[Any, list]]:
        raise NotImplementedError

    def _publish(self, name: str, blob: bytes) -> None:
        raise NotImplementedError

    def cleanup(self) -> None:
        pass


class MemFS(FS):
    def __init__(self, path: str):
        self.path = path
        with _MEM_LOCK:
            self._files = _MEM.setdefault(path, {})

    def list(self, pattern: str) -> List[str]:
        rx = re.compile(pattern)
        with _MEM_LOCK:
            return sorted(n for n in self._files if rx.match(n))

    def remove(self, name: str) -> None:
        with _MEM_LOCK:
            self._files.pop(name, None)

    def records(self, name: str):
        with _MEM_LOCK:
            blob = self._files[name]
        return read_records(io.BytesIO(blob))

    def _publish(self, name: str, blob: bytes) -> None:
        with _MEM_LOCK:
            self._files[name] = blob

    def cleanup(self) -> None:
        with _MEM_LOCK:
            _MEM.pop(self.path, None)


class SharedFS(FS):
    """Directory-backed shuffle storage (fs.lua sharedfs :119-137)."""

    def __init__(self, path: str):
        self.path = path
        os.makedirs(path, exist_ok=True)

    def _p(self, name: str) -> str:
        assert "/" not in name and ".." not in name, name
        return os.path.join(self.path, name)

    def list(self, pattern: str) -> List[str]:
        rx = re.compile(pattern)
        try:
            names = os.listdir(self.path)
        except FileNotFoundError:
            return []
        return sorted(n for n in names if rx.match(n) and not n.endswith(".tmp"))

    def remove(self, name: str) -> None:
        try:
            os.unlink(self._p(name))
        except FileNotFoundError:
            pass

    def records(self, name: str):
        fh = open(self._p(name), "rb")
        try:
            yield from read_records(fh)
        finally:
            fh.close()

    def _publish(self, name: str, blob: bytes) -> None:
        # tmpfile + rename = atomic, idempotent republish on retry
        fd, tmp = tempfile.mkstemp(dir=self.path, suffix=".tmp")
        with os.fdopen(fd, "wb") as fh:
            fh.write(blob)
        os.replace(tmp, self._p(name))

    def cleanup(self) -> None:
        try:
            for n in os.listdir(self.path):
                try:
                    os.unlink(os.path.join(self.path, n))
                except OSError:
                    pass
            os.rmdir(self.path)
        except OSError:
            pass


def router(storage: str, path: str = "") -> FS:
    """Parse "mem[:path]" / "shared[:path]" into a backend (fs.lua:185-208,
    get_storage_from utils.lua:273-285)."""
    kind, _, spath = storage.partition(":")
    spath = path or spath
    if kind == "mem":
        return MemFS(spath or "default")
    if kind == "shared":
        if not spath:
            spath = os.path.join(tempfile.gettempdir(), "mr_amd_shared")
        return SharedFS(spath)
    raise ValueError(f"unknown storage kind {kind!r} (expected mem|shared)")
