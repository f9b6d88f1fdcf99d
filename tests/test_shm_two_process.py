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


@pytest.mark.timeout(180)
def test_remote_cli_end_to_end():
    """The harness.remote CLI: --serve in one process, --connect in
    another, remote verification verdict in the client output."""
    import os
    import re
    import subprocess
    import sys
    import time

    ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    srv = subprocess.Popen(
        [sys.executable, "-m", "rocnrdma_amd.harness.remote", "--serve",
         "--region", str(REGION)],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True,
        env=env, cwd=ROOT)
    try:
        line = srv.stdout.readline()
        m = re.search(r"OOB port (\d+)", line)
        assert m, line
        port = m.group(1)
        out = subprocess.run(
            [sys.executable, "-m", "rocnrdma_amd.harness.remote",
             "--connect", f"127.0.0.1:{port}", "--region", str(REGION),
             "--msg", str(MSG), "--secs", "0.3"],
            capture_output=True, text=True, env=env, cwd=ROOT,
            timeout=120)
        assert out.returncode == 0, out.stderr
        assert "'remote_verify_bad': 0" in out.stdout
    finally:
        deadline = time.time() + 20
        while srv.poll() is None and time.time() < deadline:
            time.sleep(0.2)
        if srv.poll() is None:
            srv.kill()
