"""Hash-consed immutable composite keys.

Parity with mapreduce/tuple.lua: the reference interns tuples into 2^18 weak
buckets so composite keys compare by reference, gives them length-then-
lexicographic ordering (:183-201), a Jenkins one-at-a-time hash (:121-140)
and a tuple{...} string form.  In Python, tuples are already immutable and
hash/compare by value; what we add is (a) interning so repeated composite
keys share one object (memory + id-compare), (b) the same length-first
ordering, (c) fnv1a32/jenkins hashes used by default partition functions so
CPU and HIP partitioning agree bit-for-bit (mapreduce_amd/ops/hip/tokenize.hip
implements the same FNV-1a).
"""

from __future__ import annotations

import weakref
from typing import Any

PyTuple = tuple

_MASK32 = 0xFFFFFFFF
_MASK64 = 0xFFFFFFFFFFFFFFFF

FNV32_PRIME = 16777619
FNV32_OFFSET = 2166136261
FNV64_PRIME = 0x100000001B3
FNV64_OFFSET = 0xCBF29CE484222325


def _key_bytes(v: Any) -> bytes:
    if isinstance(v, bytes):
        return v
    if isinstance(v, str):
        return v.encode("utf-8")
    if isinstance(v, bool):
        return b"\x01" if v else b"\x00"
    if isinstance(v, int):
        return v.to_bytes(8, "little", signed=True)
    if isinstance(v, float):
        import struct

        return struct.pack("<d", v)
    raise TypeError(f"unhashable key component: {type(v)!r}")


def fnv1a32(data: Any) -> int:
    """32-bit FNV-1a over the byte image of a key.

    Matches the WordCount example's partition hash
    (examples/WordCount/partitionfn.lua:2-16): h = (h*prime) mod 2^32 then
    xor byte — note the reference multiplies BEFORE xor, so we keep that
    exact order for bit-parity between CPU, example, and HIP kernels.
    """
    h = FNV32_OFFSET
    for b in _key_bytes(data):
        h = (h * FNV32_PRIME) & _MASK32
        h ^= b
    return h


def fnv1a64(data: Any) -> int:
    """64-bit FNV-1a (standard xor-then-multiply) — the GPU key-hash
    (SURVEY.md K2/K3); must match ops/hip/tokenize.hip fnv1a64()."""
    h = FNV64_OFFSET
    for b in _key_bytes(data):
        h ^= b
        h = (h * FNV64_PRIME) & _MASK64
    return h


def wordhash64(data: Any) -> int:
    """Chunked 64-bit word hash: one xor-multiply per 8 little-endian
    bytes (zero-padded) + a length fold.  The GPU engine's dictionary hash
    (tokenize_v6): 6x fewer dependent multiplies than byte-serial FNV-1a
    and provably collision-free for distinct words of <= 8 bytes (the
    multiply by an odd prime is bijective).  Must match whash_* in
    ops/hip/common.h."""
    b = _key_bytes(data)
    h = FNV64_OFFSET
    for i in range(0, len(b), 8):
        chunk = int.from_bytes(b[i:i + 8], "little")
        h = ((h ^ chunk) * FNV64_PRIME) & _MASK64
    return ((h ^ len(b)) * FNV64_PRIME) & _MASK64


def splitmix64(x: int) -> int:
    """SplitMix64 finalizer — the doc-id mixer for composite
    (word, doc) keys in the GPU inverted index (must match the torch
    implementation in gpu/inverted_index.py)."""
    z = (x + 0x9E3779B97F4A7C15) & _MASK64
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & _MASK64
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & _MASK64
    return (z ^ (z >> 31)) & _MASK64


def jenkins_oaat(data: Any) -> int:
    """Jenkins one-at-a-time 32-bit hash (tuple.lua:121-140)."""
    h = 0
    for b in _key_bytes(data):
        h = (h + b) & _MASK32
        h = (h + ((h << 10) & _MASK32)) & _MASK32
        h ^= h >> 6
    h = (h + ((h << 3) & _MASK32)) & _MASK32
    h ^= h >> 11
    h = (h + ((h << 15) & _MASK32)) & _MASK32
    return h


class InternedTuple:
    """Immutable interned composite key with length-first ordering
    (tuple.lua:183-201: __lt/__le compare length then lexicographic).

    A proxy over a plain tuple — mirroring the reference's proxy metatable
    (tuple.lua:250-303) and, unlike a tuple subclass, weak-referenceable so
    the intern table can hold it weakly (weak buckets, tuple.lua:73-78).
    Immutable: no attribute assignment, supports hashing, indexing,
    iteration, len; repr is ``tuple{a,b,c}`` like the reference __tostring.
    """

    __slots__ = ("_items", "_hash", "__weakref__")

    def __init__(self, items: tuple):
        object.__setattr__(self, "_items", items)
        object.__setattr__(self, "_hash", hash(items))

    def __setattr__(self, *a):
        raise AttributeError("InternedTuple is immutable (tuple.lua:302)")

    def __hash__(self):
        return self._hash

    def __len__(self):
        return len(self._items)

    def __iter__(self):
        return iter(self._items)

    def __getitem__(self, i):
        return self._items[i]

    @staticmethod
    def _raw(other):
        return other._items if isinstance(other, InternedTuple) else other

    def __eq__(self, other):
        return self._items == self._raw(other)

    def __ne__(self, other):
        return self._items != self._raw(other)

    def _cmp_key(self):
        return (len(self._items), self._items)

    def __lt__(self, other):
        o = self._raw(other)
        return (len(self._items), self._items) < (len(o), o)

    def __le__(self, other):
        o = self._raw(other)
        return (len(self._items), self._items) <= (len(o), o)

    def __gt__(self, other):
        o = self._raw(other)
        return (len(self._items), self._items) > (len(o), o)

    def __ge__(self, other):
        o = self._raw(other)
        return (len(self._items), self._items) >= (len(o), o)

    def __repr__(self):
        return "tuple{" + ",".join(repr(x) for x in self._items) + "}"

    def __reduce__(self):
        # pickles re-intern on load so cross-process identity semantics hold
        return (tuple_, tuple(self._items))


_intern: "weakref.WeakValueDictionary[PyTuple, InternedTuple]" = weakref.WeakValueDictionary()


def tuple_(*args: Any) -> InternedTuple:
    """Interning constructor (tuple.lua:250-303).

    ``tuple_(a, b) is tuple_(a, b)`` holds while any reference is alive;
    buckets are weak so unused tuples are collected (the reference compacts
    weak buckets at MAX_BUCKET_HOLES_RATIO; Python's WeakValueDictionary
    does this for us).  Nested tuples/lists are interned recursively.
    """
    norm = PyTuple(
        tuple_(*a) if isinstance(a, (tuple, list)) else a for a in args
    )
    got = _intern.get(norm)
    if got is not None:
        return got
    t = InternedTuple(norm)
    _intern[norm] = t
    return t


def stats() -> dict:
    """Interning stats (tuple.lua:332-343)."""
    return {"size": len(_intern)}
