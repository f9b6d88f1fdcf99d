"""Model-based randomized transport testing (hypothesis): arbitrary
post sequences against a numpy model of the region must agree — the
offset/slot arithmetic every backend shares is the thing most likely to
hide an off-by-one."""
import numpy as np
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from rocnrdma_amd.transport import get_transport

GEOMS = st.sampled_from([
    (64, 1024), (128, 1024), (256, 4096), (1024, 8192), (512, 512),
])


@given(GEOMS, st.integers(1, 16),
       st.lists(st.integers(0, 10_000), min_size=1, max_size=60))
@settings(max_examples=60, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
def test_fake_write_matches_model(geom, inflight, posts):
    msg, region = geom
    tp = get_transport("fake", msg_bytes=msg, region_bytes=region,
                       inflight=inflight)
    model = np.zeros(region, dtype=np.uint8)
    # give every slot distinctive content
    for s in range(tp.inflight):
        tp.staging[s][:] = (s * 37 + 11) % 256
    for i in posts:
        tp.post(i)
        tp.flush()
        slot = i % tp.inflight
        off = (i % tp.msgs_per_region) * msg
        model[off:off + msg] = (slot * 37 + 11) % 256
    assert (tp.region == model).all()


@given(GEOMS, st.lists(st.integers(0, 10_000), min_size=1, max_size=40))
@settings(max_examples=40, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
def test_fake_read_matches_model(geom, posts):
    msg, region = geom
    tp = get_transport("fake", msg_bytes=msg, region_bytes=region,
                       direction="read")
    rng = np.random.default_rng(1)
    content = rng.integers(0, 256, region, dtype=np.uint8)
    tp.region[:] = content
    for i in posts:
        tp.post(i)
        tp.flush()
        slot = i % tp.inflight
        off = (i % tp.msgs_per_region) * msg
        assert (tp.staging[slot] == content[off:off + msg]).all()


@given(GEOMS, st.integers(0, 5000), st.integers(1, 64))
@settings(max_examples=40, deadline=None)
def test_post_many_equals_posts(geom, start, n):
    msg, region = geom
    tp1 = get_transport("fake", msg_bytes=msg, region_bytes=region)
    tp2 = get_transport("fake", msg_bytes=msg, region_bytes=region)
    for s in range(tp1.inflight):
        tp1.staging[s][:] = s + 1
        tp2.staging[s][:] = s + 1
    for i in range(start, start + n):
        tp1.post(i)
    tp1.flush()
    tp2.post_many(start, n)
    tp2.flush()
    assert (tp1.region == tp2.region).all()
