"""Two-process one-sided transport: a passive target process registers
memory; the initiator writes into it and the TARGET verifies — the
client/server shape of ib_write_bw, testable with no HCA."""
import multiprocessing as mp

import pytest

from rocnrdma_amd.transport.shm import ShmInitiatorTransport, target_serve

REGION = 4 << 20
MSG = 64 << 10


@pytest.fixture
def target():
    parent, child = mp.Pipe()
    proc = mp.Process(target=target_serve, args=(child, REGION),
                      daemon=True)
    proc.start()
    port = parent.recv()
    yield port
    proc.join(timeout=30)
    if proc.is_alive():
        proc.terminate()


@pytest.mark.timeout(120)
def test_one_sided_write_remote_verify(target):
    tp = ShmInitiatorTransport(msg_bytes=MSG, region_bytes=REGION,
                               host="127.0.0.1", port=target)
    # bandwidth-phase posts (content arbitrary), then integrity: the
    # REMOTE process verifies its own memory
    tp.post_many(0, tp.msgs_per_region * 2)
    tp.flush()
    assert tp.integrity_check(seed=99) == 0
    tp.close()


@pytest.mark.timeout(120)
def test_one_sided_read(target):
    tp = ShmInitiatorTransport(msg_bytes=MSG, region_bytes=REGION,
                               host="127.0.0.1", port=target,
                               direction="read")
    assert tp.integrity_check(seed=123) == 0
    tp.close()


@pytest.mark.timeout(120)
def test_remote_verify_detects_wrong_seed(target):
    tp = ShmInitiatorTransport(msg_bytes=MSG, region_bytes=REGION,
                               host="127.0.0.1", port=target)
    assert tp.integrity_check(seed=5) == 0
    # ask the target to verify against a DIFFERENT seed: must mismatch
    tp.oob.send({"op": "verify", "seed": 6})
    assert tp.oob.recv()["bad"] > 0
    tp.close()
