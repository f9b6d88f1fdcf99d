"""Property-based tests (hypothesis) for the pure-logic pieces the
whole stack leans on."""
import zlib

import numpy as np
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

PAGE_SETTINGS = settings(
    max_examples=20, deadline=None,
    suppress_health_check=[HealthCheck.large_base_example])

from rocnrdma_amd.utils import pattern
from rocnrdma_amd.utils.topology import parse_cpulist

from tests.test_pattern import _gf2_times, _shift_matrices


@given(st.integers(0, 2**64 - 1), st.integers(0, 1000),
       st.integers(1, 256))
@settings(max_examples=50, deadline=None)
def test_pattern_windows_compose(seed, start, count):
    """Any window of the pattern equals the same slice of a larger
    window (the property integrity checks rely on)."""
    w = pattern.splitmix64_words(seed, start, count)
    full = pattern.splitmix64_words(seed, 0, start + count)
    assert (w == full[start:start + count]).all()


@given(st.binary(min_size=4096, max_size=4096))
@PAGE_SETTINGS
def test_lane_combine_any_page(page):
    """The GPU CRC kernel's combine algebra holds for arbitrary data,
    not just random-looking data."""
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


@given(st.binary(min_size=4096, max_size=4096))
@PAGE_SETTINGS
def test_tree_combine_equals_flat(page):
    """The log-tree combine the tuned kernel uses is algebraically the
    flat XOR-of-shifts (shift is linear over XOR)."""
    mats = _shift_matrices()

    def shift_level(c, k):
        return _gf2_times(mats[k], c)

    crcs = [zlib.crc32(page[l * 64:(l + 1) * 64]) & 0xFFFFFFFF
            for l in range(64)]
    # tree: level k merges spans of 64B << k
    vals = list(crcs)
    for k in range(6):
        nxt = []
        for j in range(0, len(vals), 2):
            nxt.append(shift_level(vals[j], k) ^ vals[j + 1])
        vals = nxt
    assert vals[0] == (zlib.crc32(page) & 0xFFFFFFFF)


@given(st.lists(st.tuples(st.integers(0, 1000), st.integers(0, 100)),
                max_size=30))
@settings(max_examples=100, deadline=None)
def test_parse_cpulist_roundtrip(ranges):
    cpus = sorted({c for lo, n in ranges for c in range(lo, lo + n)})
    # build a cpulist string with ranges
    parts = []
    i = 0
    while i < len(cpus):
        j = i
        while j + 1 < len(cpus) and cpus[j + 1] == cpus[j] + 1:
            j += 1
        parts.append(str(cpus[i]) if i == j else f"{cpus[i]}-{cpus[j]}")
        i = j + 1
    assert parse_cpulist(",".join(parts)) == cpus


@given(st.integers(0, 2**64 - 1), st.integers(1, 4096))
@settings(max_examples=30, deadline=None)
def test_fill_reference_is_uint8_view(seed, nwords):
    buf = pattern.fill_reference(nwords * 8, seed)
    assert buf.dtype == np.uint8
    assert (buf.view(np.uint64) ==
            pattern.splitmix64_words(seed, 0, nwords)).all()
