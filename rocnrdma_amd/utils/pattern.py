"""CPU reference implementations of the GPU payload kernels.

Bit-identical to rocnrdma_amd/ops/csrc/p2p_kernels.hip — the GPU
numerics tests compare the HIP kernels against these.
"""
from __future__ import annotations

import zlib

import numpy as np

_GOLDEN = np.uint64(0x9E3779B97F4A7C15)
_M1 = np.uint64(0xBF58476D1CE4E5B9)
_M2 = np.uint64(0x94D049BB133111EB)

PAGE = 4096


def splitmix64_words(seed: int, start: int, count: int) -> np.ndarray:
    """Pattern words i in [start, start+count): mix(seed + (i+1)*GOLDEN)."""
    with np.errstate(over="ignore"):
        i = np.arange(start + 1, start + count + 1, dtype=np.uint64)
        x = np.uint64(seed & 0xFFFFFFFFFFFFFFFF) + i * _GOLDEN
        x = (x ^ (x >> np.uint64(30))) * _M1
        x = (x ^ (x >> np.uint64(27))) * _M2
        return x ^ (x >> np.uint64(31))


def fill_reference(nbytes: int, seed: int) -> np.ndarray:
    """The full pattern buffer as uint8 (nbytes % 8 == 0)."""
    assert nbytes % 8 == 0
    words = splitmix64_words(seed, 0, nbytes // 8)
    return words.view(np.uint8)


def crc32_pages_reference(data: bytes | np.ndarray) -> np.ndarray:
    """zlib.crc32 of each 4 KiB page, as uint32 array."""
    buf = np.asarray(data, dtype=np.uint8).tobytes()
    assert len(buf) % PAGE == 0
    return np.array(
        [zlib.crc32(buf[i : i + PAGE]) for i in range(0, len(buf), PAGE)],
        dtype=np.uint32,
    )
