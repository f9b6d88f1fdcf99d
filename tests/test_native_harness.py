"""Builds and exercises the native C++ harness (rocp2p_bw) on the fake
backend — CPU tier; its hip/verbs backends run on GPU/HCA hosts."""
import json
import os
import shutil
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HARNESS = os.path.join(ROOT, "harness")
BIN = os.path.join(HARNESS, "build", "rocp2p_bw")


@pytest.fixture(scope="module")
def built():
    if shutil.which("make") is None or shutil.which("hipcc") is None:
        pytest.skip("no toolchain")
    subprocess.run(["make", "-C", HARNESS], check=True, capture_output=True,
                   text=True, timeout=900)
    return BIN


@pytest.mark.timeout(900)
def test_fake_write_json(built):
    out = subprocess.run(
        [built, "--transport", "fake", "--msg", "65536", "--region",
         "4194304", "--secs", "0.2", "--json"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    r = json.loads(out.stdout.strip())
    assert r["transport"] == "fake"
    assert r["gbps"] > 0
    assert r["integrity"] == "ok"


@pytest.mark.timeout(300)
def test_fake_read_direction(built):
    out = subprocess.run(
        [built, "--transport", "fake", "--msg", "4096", "--region",
         "1048576", "--secs", "0.1", "--dir", "read", "--json"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    r = json.loads(out.stdout.strip())
    assert r["direction"] == "read"
    assert r["integrity"] == "ok"


@pytest.mark.timeout(300)
def test_sweep_mode(built):
    out = subprocess.run(
        [built, "--transport", "fake", "--region", "8388608", "--secs",
         "0.05", "--sweep", "--json"],
        capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr
    lines = [json.loads(l) for l in out.stdout.strip().splitlines()]
    assert len(lines) == 5  # 4K..64M
    assert all(r["integrity"] == "ok" for r in lines)


@pytest.mark.timeout(300)
def test_verbs_unavailable_is_actionable(built):
    out = subprocess.run(
        [built, "--transport", "verbs", "--msg", "4096", "--region",
         "1048576", "--secs", "0.05"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 3
    assert "verbs" in out.stderr.lower()


@pytest.mark.timeout(300)
def test_lat_mode(built):
    out = subprocess.run(
        [built, "--transport", "fake", "--msg", "4096", "--region",
         "1048576", "--lat", "500", "--json"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    r = json.loads(out.stdout.strip())
    assert r["mode"] == "lat"
    assert 0 < r["us_min"] <= r["us_p50"] <= r["us_p99"] <= r["us_max"]
    assert r["integrity"] == "ok"


FAKEVERBS = os.path.join(HARNESS, "build", "rocp2p_bw_fakeverbs")


@pytest.fixture(scope="module")
def built_fakeverbs(built):
    assert os.path.exists(FAKEVERBS)
    return FAKEVERBS


@pytest.mark.parametrize("link", ["ib", "eth"])
@pytest.mark.parametrize("direction", ["write", "read"])
@pytest.mark.timeout(300)
def test_verbs_backend_against_fake_layer(built_fakeverbs, link, direction):
    """The REAL verbs backend (QP bring-up masks, MR access flags, WR
    posting, CQ drain) validated by the strict in-process fake, on both
    link-layer connect paths."""
    env = dict(os.environ, FAKE_VERBS_LINK=link)
    out = subprocess.run(
        [built_fakeverbs, "--transport", "verbs", "--mr", "host", "--msg",
         "65536", "--region", "2097152", "--secs", "0.1", "--dir",
         direction, "--json"],
        capture_output=True, text=True, timeout=120, env=env)
    assert out.returncode == 0, out.stderr
    r = json.loads(out.stdout.strip())
    assert r["transport"] == "verbs"
    assert r["integrity"] == "ok"


@pytest.mark.timeout(300)
def test_verbs_backend_lat_against_fake_layer(built_fakeverbs):
    out = subprocess.run(
        [built_fakeverbs, "--transport", "verbs", "--mr", "host", "--msg",
         "4096", "--region", "1048576", "--lat", "300", "--json"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    r = json.loads(out.stdout.strip())
    assert r["mode"] == "lat" and r["integrity"] == "ok"


@pytest.mark.parametrize("link", ["ib", "eth"])
@pytest.mark.timeout(300)
def test_verbs_client_server_remote_selftest(built_fakeverbs, link):
    """Full ib_write_bw client/server shape through the real verbs
    backend: TCP OOB exchange, remote QP bring-up, one-sided writes
    into the server's registered region, REMOTE integrity verdict."""
    env = dict(os.environ, FAKE_VERBS_LINK=link)
    out = subprocess.run(
        [built_fakeverbs, "--transport", "verbs", "--mr", "host",
         "--msg", "65536", "--region", "2097152", "--remote-selftest",
         "--secs", "0.1"],
        capture_output=True, text=True, timeout=120, env=env)
    assert out.returncode == 0, out.stderr
    r = json.loads(out.stdout.strip())
    assert r["mode"] == "remote-selftest"
    assert r["remote_integrity"] == "ok"


@pytest.mark.timeout(300)
def test_verbs_client_server_read_direction(built_fakeverbs):
    out = subprocess.run(
        [built_fakeverbs, "--transport", "verbs", "--mr", "host",
         "--msg", "4096", "--region", "1048576", "--remote-selftest",
         "--secs", "0.1", "--dir", "read"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    assert json.loads(out.stdout.strip())["remote_integrity"] == "ok"


@pytest.mark.parametrize("direction", ["write", "read"])
@pytest.mark.timeout(300)
def test_verbs_dmabuf_mr_end_to_end(built_fakeverbs, direction):
    """the ibv_reg_dmabuf_mr path must EXECUTE, not just
    compile: the region is an fd-exported buffer (memfd on CPU; real
    VRAM dmabuf on a GPU box), the fake core mmaps the fd the way an
    HCA would DMA the exporter's pages, and one-sided ops move real
    bytes through that mapping with integrity checked."""
    out = subprocess.run(
        [built_fakeverbs, "--transport", "verbs", "--mr", "dmabuf",
         "--msg", "65536", "--region", "4194304", "--secs", "0.1",
         "--dir", direction, "--json"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stdout + out.stderr
    r = json.loads(out.stdout.strip())
    assert r["integrity"] == "ok"


@pytest.mark.parametrize("chain", ["1", "7", "16"])
@pytest.mark.timeout(300)
def test_verbs_wr_chaining(built_fakeverbs, chain):
    """chained posting (one doorbell per up-to-chain WRs,
    selective signaling, chain retired per completion) must preserve
    exact delivery for every chain length, including one that does not
    divide the message count."""
    out = subprocess.run(
        [built_fakeverbs, "--transport", "verbs", "--msg", "4096",
         "--region", "1048576", "--secs", "0.1", "--chain", chain,
         "--json"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stdout + out.stderr
    r = json.loads(out.stdout.strip())
    assert r["integrity"] == "ok"
    assert r["msgs_per_s"] > 0


FULLSTACK = os.path.join(HARNESS, "build", "rocp2p_bw_fullstack")


@pytest.fixture(scope="module")
def built_fullstack(built):
    assert os.path.exists(FULLSTACK)
    return FULLSTACK


@pytest.mark.parametrize("direction", ["write", "read"])
@pytest.mark.timeout(300)
def test_full_stack_peer_mr(built_fullstack, direction):
    """The reference's COMPLETE L5->L0 flow in one
    process: ibv_reg_mr on a GPU VA -> IB-core peer probe -> the REAL
    rocp2p bridge (acquire/get_pages/dma_map against the fake KFD) ->
    one-sided DMA against the bridge's device-mapped sg table ->
    payload verified in 'VRAM'.  No layer mocked out of the control
    path: the bridge code on this hot path is the same file kbuild
    compiles (module/bridge/rocp2p_main.c)."""
    out = subprocess.run(
        [built_fullstack, "--transport", "verbs", "--mr", "peer",
         "--msg", "65536", "--region", "8388608", "--secs", "0.1",
         "--dir", direction, "--json"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stdout + out.stderr
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["integrity"] == "ok"
    # the real bridge registered (its init line proves dispatch)
    assert "PeerDirect client 'rocp2p'" in out.stderr


@pytest.mark.timeout(300)
def test_full_stack_peer_mr_client_server(built_fullstack):
    """ib_write_bw server/client shape with the server's region
    registered through the real bridge (peer MR), client RDMA-writes
    over TCP-bootstrapped QPs, REMOTE integrity verification."""
    out = subprocess.run(
        [built_fullstack, "--remote-selftest", "--mr", "peer",
         "--msg", "65536", "--region", "8388608", "--secs", "0.1"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stdout + out.stderr
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["remote_integrity"] == "ok"


@pytest.mark.timeout(300)
def test_full_stack_peer_mr_chained(built_fullstack):
    """Chained posting against a bridge-registered MR keeps exact
    delivery."""
    out = subprocess.run(
        [built_fullstack, "--transport", "verbs", "--mr", "peer",
         "--msg", "4096", "--region", "4194304", "--secs", "0.1",
         "--chain", "16", "--json"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stdout + out.stderr
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["integrity"] == "ok"


def _two_process_fabric(binary, mr, tmp_path, msg="65536",
                        region="8388608"):
    import re
    import time

    env = dict(os.environ, FAKE_VERBS_SHM=f"/rocnr_fab_{os.getpid()}")
    srv = subprocess.Popen(
        [binary, "--serve", "0", "--mr", mr, "--region", region],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True,
        env=env)
    try:
        port = None
        t0 = time.time()
        while time.time() - t0 < 30 and port is None:
            line = srv.stdout.readline()
            m = re.search(r"listening on port (\d+)", line)
            if m:
                port = int(m.group(1))
        assert port, "server never announced"
        out = subprocess.run(
            [binary, "--connect", f"127.0.0.1:{port}", "--msg", msg,
             "--region", region, "--secs", "0.1"],
            capture_output=True, text=True, timeout=120, env=env)
        assert out.returncode == 0, out.stdout + out.stderr
        r = json.loads(out.stdout.strip().splitlines()[-1])
        assert r["remote_integrity"] == "ok"
        srv.wait(timeout=30)
        assert srv.returncode == 0
    finally:
        if srv.poll() is None:
            srv.kill()
        srv.communicate()
        shm = "/dev/shm" + env["FAKE_VERBS_SHM"]
        if os.path.exists(shm):
            os.unlink(shm)


@pytest.mark.timeout(300)
def test_two_process_fabric_host_mr(built_fakeverbs, tmp_path):
    """TWO real processes over TCP + the shm fabric: the client's
    RDMA WRITEs are applied by the server process's NIC-role engine
    (the server's application thread never touches the data path —
    RDMA semantics); remote verification runs in the server."""
    _two_process_fabric(built_fakeverbs, "host", tmp_path)


@pytest.mark.timeout(300)
def test_two_process_fabric_peer_mr(built_fullstack, tmp_path):
    """Same, with the server's region registered through the REAL
    bridge (peer MR): cross-process one-sided writes land in
    bridge-pinned 'VRAM' and are verified remotely — the reference's
    deployment shape (ib_write_bw server with GPU memory) across
    genuine process boundaries."""
    _two_process_fabric(built_fullstack, "peer", tmp_path)


@pytest.mark.timeout(300)
def test_peer_revoke_through_verbs_surface(built_fullstack):
    """SURVEY §3.4 visible from L5: the producer frees memory under a
    live bridge-registered MR; the invalidation (bridge free_cb ->
    IB-core invalidate -> teardown) must make subsequent posts fail as
    remote-access errors and leave deregistration idempotent."""
    out = subprocess.run([built_fullstack, "--peer-revoke-selftest"],
                         capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stdout + out.stderr
    assert '"result":"ok"' in out.stdout
    assert "peer MR invalidated" in out.stderr


@pytest.mark.timeout(300)
def test_two_process_fabric_large_messages(built_fakeverbs, tmp_path):
    """Messages larger than the fabric's 1 MiB slot payload must be
    chunked transparently (one WR = several fabric ops) with exact
    remote delivery."""
    _two_process_fabric(built_fakeverbs, "host", tmp_path,
                        msg="4194304", region="16777216")


@pytest.mark.parametrize("n", ["2", "4"])
@pytest.mark.timeout(300)
def test_verbs_multi_qp_fanout(built_fakeverbs, n):
    """--gpus N over the verbs backend: N concurrent workers, each
    with its own QP pair/MRs/CQ in one process (the 8-GPU node fan-out
    shape at the verbs layer); integrity per worker, aggregate
    reported."""
    out = subprocess.run(
        [built_fakeverbs, "--transport", "verbs", "--gpus", n,
         "--msg", "65536", "--region", "4194304", "--secs", "0.1",
         "--json"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stdout + out.stderr
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["gpus"] == int(n)
    assert r["integrity"] == "ok"
