"""Array-backed binary min-heap with an injectable key/comparator.

Parity with mapreduce/heap.lua (push :55-70, pop :33-53, top/clear/size/empty
:29-82).  The reference uses it as the shuffle-merge hot structure; here it
only backs the host-side general-reducer merge path — the GPU tier replaces
the merge with radix sort + segmented reduce.
"""

from __future__ import annotations

from typing import Any, Callable, List, Optional


class Heap:
    __slots__ = ("_data", "_key")

    def __init__(self, key: Optional[Callable[[Any], Any]] = None):
        self._data: List[Any] = []
        self._key = key or (lambda x: x)

    def size(self) -> int:
        return len(self._data)

    def empty(self) -> bool:
        return not self._data

    def clear(self) -> None:
        self._data.clear()

    def top(self) -> Any:
        if not self._data:
            raise IndexError("top of empty heap")
        return self._data[0]

    def push(self, item: Any) -> None:
        d, key = self._data, self._key
        d.append(item)
        i = len(d) - 1
        ki = key(item)
        while i > 0:
            parent = (i - 1) >> 1
            if key(d[parent]) <= ki:
                break
            d[i] = d[parent]
            i = parent
        d[i] = item

    def pop(self) -> Any:
        d, key = self._data, self._key
        if not d:
            raise IndexError("pop from empty heap")
        out = d[0]
        last = d.pop()
        n = len(d)
        if n:
            i = 0
            klast = key(last)
            while True:
                l = 2 * i + 1
                r = l + 1
                small = i
                ksmall = klast
                if l < n:
                    kl = key(d[l])
                    if kl < ksmall:
                        small, ksmall = l, kl
                if r < n:
                    kr = key(d[r])
                    if kr < ksmall:
                        small, ksmall = r, kr
                if small == i:
                    break
                d[i] = d[small]
                i = small
            d[i] = last
        return out
