"""Distributed persistent key/value singleton.

Parity with mapreduce/persistent_table.lua: a shared config/state table
persisted in the control plane, with optimistic concurrency via a timestamp
counter (findAndModify-CAS, :41-74), a cooperative lock()/unlock() spin lock
(:113-138), read_only mode, and reserved-field checks (:95-110).  Used by
iterative training tasks to share e.g. the current model blob / finished
flag across processes (APRIL-ANN common.lua:57-77).
"""

from __future__ import annotations

import time
from typing import Any, Optional

from .parallel.coord import Coordinator, connect

_RESERVED = {"_id", "timestamp", "locked"}


class PersistentTable:
    def __init__(self, name: str, cnn_string: str = "local", db: str = "mr",
                 coord: Optional[Coordinator] = None, read_only: bool = False):
        object.__setattr__(self, "_coord",
                           coord or connect(cnn_string, db, listen=False))
        object.__setattr__(self, "_key", f"singletons/{name}")
        object.__setattr__(self, "_read_only", read_only)
        object.__setattr__(self, "_doc", None)
        object.__setattr__(self, "_raw", None)
        object.__setattr__(self, "_dirty", {})
        self.update()

    # -- sync ---------------------------------------------------------------
    def update(self) -> None:
        """Push dirty fields (CAS with timestamp bump) then pull
        (persistent_table.lua:41-74)."""
        coord: Coordinator = self._coord
        for _ in range(1024):
            doc, raw = coord.get_doc(self._key)
            if doc is None:
                doc = {"_id": self._key, "timestamp": 0, "locked": 0}
                raw = None
            if not self._dirty:
                object.__setattr__(self, "_doc", doc)
                object.__setattr__(self, "_raw", raw)
                return
            if self._read_only:
                raise PermissionError("read_only persistent_table")
            new = dict(doc)
            new.update(self._dirty)
            new["timestamp"] = doc["timestamp"] + 1
            if coord.cas_doc(self._key, raw, new):
                self._dirty.clear()
                object.__setattr__(self, "_doc", new)
                object.__setattr__(self, "_raw", None)
                # refresh raw for future CAS
                _, raw2 = coord.get_doc(self._key)
                object.__setattr__(self, "_raw", raw2)
                return
        raise RuntimeError("persistent_table CAS livelock")

    def drop(self) -> None:
        """persistent_table.lua:77-93."""
        self._coord.delete_doc(self._key)
        object.__setattr__(self, "_doc", None)
        self._dirty.clear()

    # -- lock ----------------------------------------------------------------
    def lock(self, timeout: float = 60.0) -> None:
        """Cooperative spin lock on the 'locked' field
        (persistent_table.lua:113-138)."""
        coord: Coordinator = self._coord
        deadline = time.time() + timeout
        while time.time() < deadline:
            doc, raw = coord.get_doc(self._key)
            if doc is None:
                doc = {"_id": self._key, "timestamp": 0, "locked": 0}
                raw = None
            if not doc.get("locked"):
                new = dict(doc)
                new["locked"] = 1
                new["timestamp"] = doc["timestamp"] + 1
                if coord.cas_doc(self._key, raw, new):
                    return
            time.sleep(0.01)  # reference spins at 0.1 s
        raise TimeoutError("persistent_table lock timeout")

    def unlock(self) -> None:
        """persistent_table.lua:140-161."""
        coord: Coordinator = self._coord
        while True:
            doc, raw = coord.get_doc(self._key)
            if doc is None or not doc.get("locked"):
                return
            new = dict(doc)
            new["locked"] = 0
            new["timestamp"] = doc["timestamp"] + 1
            if coord.cas_doc(self._key, raw, new):
                return

    # -- dict-style access (proxy __index/__newindex, :176-252) --------------
    def __getattr__(self, name: str) -> Any:
        if name.startswith("_"):
            raise AttributeError(name)
        if name in self._dirty:
            return self._dirty[name]
        doc = self._doc or {}
        return doc.get(name)

    def __setattr__(self, name: str, value: Any) -> None:
        self.set(name, value)

    def get(self, name: str, default: Any = None) -> Any:
        v = self.__getattr__(name)
        return default if v is None else v

    def set(self, name: str, value: Any) -> None:
        if name in _RESERVED:
            raise KeyError(f"reserved field {name!r} "
                           "(persistent_table.lua:95-110)")
        if self._read_only:
            raise PermissionError("read_only persistent_table")
        self._dirty[name] = value

    def __repr__(self) -> str:
        d = {k: v for k, v in (self._doc or {}).items()
             if k not in _RESERVED}
        d.update(self._dirty)
        return f"PersistentTable({d!r})"
