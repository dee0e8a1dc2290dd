"""Deterministic, seed-independent hash for stable partitioning across
interpreter runs — the role of dpark/portable_hash.pyx [U] (SURVEY.md §2
row 6, the only native component in reference dpark).

Python's str hash is randomized per process (PYTHONHASHSEED), so a
portable hash must avoid hash() for str/bytes.  Int hashing matches
CPython's (hash(n) = n mod 2^61-1, -1 -> -2), which IS seed-independent;
str/bytes use FNV-1a 64-bit here (the exact str algorithm differs from
the reference's [U?] — partition LAYOUT need not match, SURVEY.md
§8(a2): collected results are partition-independent).

The GPU engine uses a splitmix64 digit of the int64 key instead
(dpark_amd/csrc); same design freedom.
"""

from __future__ import annotations

_M61 = (1 << 61) - 1
_FNV_OFF = 0xCBF29CE484222325
_FNV_PRIME = 0x100000001B3
_M64 = (1 << 64) - 1


def portable_hash(obj) -> int:
    if obj is None:
        return 0
    if isinstance(obj, bool):
        return int(obj)
    if isinstance(obj, int):
        h = obj % _M61 if obj >= 0 else -((-obj) % _M61)
        return -2 if h == -1 else h
    if isinstance(obj, float):
        if obj == int(obj) and abs(obj) < 2 ** 63:
            return portable_hash(int(obj))
        return portable_hash(repr(obj).encode())
    if isinstance(obj, str):
        return portable_hash(obj.encode("utf-8"))
    if isinstance(obj, (bytes, bytearray)):
        h = _FNV_OFF
        for b in obj:
            h = ((h ^ b) * _FNV_PRIME) & _M64
        h &= (1 << 62) - 1  # keep positive, fits int64
        return h
    if isinstance(obj, tuple):
        h = 0x345678
        for item in obj:
            h = ((h ^ portable_hash(item)) * 1000003) & _M64
        return h & ((1 << 62) - 1)
    raise TypeError(f"portable_hash: unhashable portable type {type(obj)}")
