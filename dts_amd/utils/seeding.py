"""Deterministic seed derivation.

Python's hash() is randomized per process for str (PYTHONHASHSEED), so
seeds derived with it change between runs. stable_seed gives the same
31-bit value for the same parts in every process.
"""

import zlib


def stable_seed(*parts) -> int:
    return zlib.crc32(repr(parts).encode("utf-8")) & 0x7FFFFFFF
