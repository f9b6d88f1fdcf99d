"""CPU tests of the payload-pattern / CRC reference implementations and
of the combine algebra the GPU CRC kernel relies on."""
import zlib

import numpy as np
import pytest

from rocnrdma_amd.utils import pattern

POLY = 0xEDB88320


def test_splitmix_deterministic_and_seeded():
    a = pattern.splitmix64_words(1, 0, 1024)
    b = pattern.splitmix64_words(1, 0, 1024)
    c = pattern.splitmix64_words(2, 0, 1024)
    assert (a == b).all()
    assert (a != c).any()
    # windowing consistency: words [10:20) equal slice of [0:30)
    w = pattern.splitmix64_words(1, 10, 10)
    assert (w == pattern.splitmix64_words(1, 0, 30)[10:20]).all()


def test_splitmix_bit_quality():
    w = pattern.splitmix64_words(99, 0, 1 << 14)
    bits = np.unpackbits(w.view(np.uint8))
    density = bits.mean()
    assert 0.49 < density < 0.51


def test_fill_reference_shape():
    buf = pattern.fill_reference(4096, 5)
    assert buf.dtype == np.uint8 and buf.size == 4096
    assert (buf.view(np.uint64) == pattern.splitmix64_words(5, 0, 512)).all()


def test_crc_reference_matches_zlib():
    rng = np.random.default_rng(0)
    data = rng.integers(0, 256, 3 * 4096, dtype=np.uint8)
    crcs = pattern.crc32_pages_reference(data)
    for p in range(3):
        assert crcs[p] == (zlib.crc32(data[p * 4096:(p + 1) * 4096].tobytes())
                           & 0xFFFFFFFF)


# ---- combine algebra used by the GPU kernel (64 lanes x 64 B) ----
def _gf2_times(mat, vec):
    s = 0
    i = 0
    while vec:
        if vec & 1:
            s ^= mat[i]
        vec >>= 1
        i += 1
    return s


def _gf2_square(mat):
    return [_gf2_times(mat, mat[i]) for i in range(32)]


def _shift_matrices():
    m = [POLY] + [1 << (i - 1) for i in range(1, 32)]  # 1 bit
    for _ in range(3):
        m = _gf2_square(m)  # -> 1 byte
    for _ in range(6):
        m = _gf2_square(m)  # -> 64 bytes
    mats = []
    for _ in range(6):
        mats.append(m)
        m = _gf2_square(m)
    return mats  # M(64B << k), k = 0..5


def test_lane_combine_identity_matches_zlib():
    rng = np.random.default_rng(7)
    page = rng.integers(0, 256, 4096, dtype=np.uint8).tobytes()
    mats = _shift_matrices()
    acc = 0
    for lane in range(64):
        c = zlib.crc32(page[lane * 64:(lane + 1) * 64]) & 0xFFFFFFFF
        tail = 63 - lane
        for k in range(6):
            if (tail >> k) & 1:
                c = _gf2_times(mats[k], c)
        acc ^= c
    assert acc == (zlib.crc32(page) & 0xFFFFFFFF)


@pytest.mark.parametrize("nbytes", [8, 4096, 65536])
def test_pattern_is_windowable(nbytes):
    full = pattern.fill_reference(nbytes, 3)
    half = pattern.fill_reference(nbytes // 2 * 2, 3)
    assert (full[: half.size] == half).all()
