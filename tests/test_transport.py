"""CPU tests of the transport layer via the fake (loopback) backend —
BASELINE config 1."""
import numpy as np
import pytest

from rocnrdma_amd.transport import available_transports, get_transport
from rocnrdma_amd.utils import pattern


def test_available_contains_fake():
    assert "fake" in available_transports()


def test_bad_geometry_rejected():
    with pytest.raises(ValueError):
        get_transport("fake", msg_bytes=4096, region_bytes=10000)
    with pytest.raises(ValueError):
        get_transport("fake", msg_bytes=4096, region_bytes=8192,
                      direction="sideways")
    with pytest.raises(ValueError):
        get_transport("nope", msg_bytes=4096, region_bytes=8192)


@pytest.mark.parametrize("direction", ["write", "read"])
def test_fake_integrity(direction):
    tp = get_transport("fake", msg_bytes=4096, region_bytes=64 * 1024,
                       direction=direction)
    assert tp.msgs_per_region == 16
    assert tp.integrity_check(seed=42) == 0


def test_fake_write_moves_bytes():
    tp = get_transport("fake", msg_bytes=1024, region_bytes=4096)
    tp.staging[0][:] = 7
    tp.post(0)
    tp.flush()
    assert (tp.region[:1024] == 7).all()
    assert (tp.region[1024:] == 0).all()


def test_fake_detects_corruption():
    tp = get_transport("fake", msg_bytes=4096, region_bytes=16 * 4096)
    assert tp.integrity_check(seed=9) == 0
    # corrupt one destination word post-hoc: a rerun must transfer fresh
    # payloads and still pass; direct verification of a corrupted region
    # is the GPU kernels' job (test_gpu_kernels.py)
    tp.region[100] ^= 0xFF
    ref = pattern.fill_reference(tp.region_bytes, 9)
    assert int(np.count_nonzero(
        tp.region.view(np.uint64) != ref.view(np.uint64))) == 1


def test_inflight_clamped():
    tp = get_transport("fake", msg_bytes=4096, region_bytes=8192,
                       inflight=64)
    assert tp.inflight == 2


def test_python_lat_sweep():
    from rocnrdma_amd.harness.sweep import run_lat_point

    tp = get_transport("fake", msg_bytes=4096, region_bytes=65536)
    r = run_lat_point(tp, iters=200)
    assert r["mode"] == "lat"
    assert 0 < r["us_min"] <= r["us_p50"] <= r["us_p99"] <= r["us_max"]


def test_soak_fake_short():
    from rocnrdma_amd.harness.soak import run_soak

    stats = run_soak("fake", secs=1.5, region_bytes=4 << 20, seed=7)
    assert stats["cycles"] > 0
    assert stats["audits"] == stats["cycles"]
    assert stats["failures"] == 0


def test_metrics_capture_in_soak():
    from rocnrdma_amd.harness.soak import run_soak
    from rocnrdma_amd.utils.metrics import TransferMetrics

    m = TransferMetrics(port=None)
    stats = run_soak("fake", secs=1.0, region_bytes=4 << 20, seed=11,
                     metrics=m)
    text = m.render().decode()
    assert "rocp2p_bytes_total" in text
    assert "rocp2p_integrity_audits_total" in text
    assert stats["failures"] == 0


def test_available_transports_cpu_box():
    names = available_transports()
    assert "fake" in names and "shm" in names
    # no GPU in CI: sdma must NOT be offered
    import torch

    if not torch.cuda.is_available():
        assert "sdma" not in names
