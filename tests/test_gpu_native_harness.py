"""GPU tests of the native C++ harness's HIP backend (built binary
travels with the repo snapshot; rebuilt on demand if missing)."""
import json
import os
import subprocess

import pytest

torch = pytest.importorskip("torch")

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU"),
]

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(ROOT, "harness", "build", "rocp2p_bw")


@pytest.fixture(scope="module")
def built():
    if not os.path.exists(BIN):
        subprocess.run(["make", "-C", os.path.join(ROOT, "harness")],
                       check=True, capture_output=True, text=True,
                       timeout=900)
    return BIN


def run(built, *args):
    out = subprocess.run([built, "--json", *args], capture_output=True,
                         text=True, timeout=300)
    assert out.returncode == 0, out.stderr
    return [json.loads(l) for l in out.stdout.strip().splitlines()]


@pytest.mark.parametrize("engine,name", [("kernel", "hip-kernel"),
                                         ("stream", "hip-stream")])
def test_hip_engines(built, engine, name):
    rows = run(built, "--transport", "hip", "--msg", "1048576", "--region",
               "33554432", "--secs", "0.2", "--engine", engine)
    assert rows[0]["transport"] == name
    assert rows[0]["integrity"] == "ok"
    assert rows[0]["gbps"] > 5


def test_hip_4k_messages(built):
    rows = run(built, "--transport", "hip", "--msg", "4096", "--region",
               "16777216", "--secs", "0.2")
    assert rows[0]["integrity"] == "ok"
    assert rows[0]["gbps"] > 10  # doorbell-batch engine, PCIe-bound


def test_hip_read_direction(built):
    rows = run(built, "--transport", "hip", "--msg", "65536", "--region",
               "16777216", "--secs", "0.2", "--dir", "read")
    assert rows[0]["integrity"] == "ok"


FAKEVERBS = os.path.join(ROOT, "harness", "build", "rocp2p_bw_fakeverbs")


@pytest.mark.parametrize("direction", ["write", "read"])
def test_dmabuf_mr_on_real_vram(built, direction):
    """dmabuf MR path on REAL HBM: hipMalloc VRAM is exported as a
    dmabuf fd (hipMemGetHandleForAddressRange), the fake verbs core
    mmaps the fd — amdgpu's dma-buf mmap hands back the PCIe BAR
    window — and one-sided ops move real bytes over the bus into VRAM,
    verified by the on-GPU kernels.  The full ibv_reg_dmabuf_mr
    lifetime executes against genuine exporter pages."""
    if not os.path.exists(FAKEVERBS):
        pytest.skip("fakeverbs harness not built")
    # read direction: CPU loads from the uncached BAR window crawl at
    # ~50 MB/s by nature (documented in profiles/r2_dmabuf_bar.md) —
    # keep that leg small so the test stays fast, and only require
    # integrity + progress.
    region = "67108864" if direction == "write" else "16777216"
    out = subprocess.run(
        [FAKEVERBS, "--transport", "verbs", "--mr", "dmabuf",
         "--msg", "1048576", "--region", region, "--secs", "0.2",
         "--dir", direction, "--json"],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stdout + out.stderr
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["integrity"] == "ok"
    assert r["gbps"] > (1.0 if direction == "write" else 0.005)


def test_bench_verbs_peer_mr_shape_on_gpu(built):
    """bench.py --transport verbs on a GPU box without the bridge
    loaded must fail ACTIONABLY from the peer-MR path (ibv_reg_mr on a
    GPU VA has no peer client to claim it in the fake core — the fake
    mock registers host/dmabuf MRs only), not silently fall back."""
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    env["ROCNR_FORCE_VERBS"] = "1"
    env["ROCNR_VERBS_HARNESS"] = FAKEVERBS
    import sys

    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--gpus", "1",
         "--steps", "2", "--warmup", "1", "--msg-bytes", "1048576",
         "--region-bytes", "16777216", "--verbs-mr", "dmabuf"],
        capture_output=True, text=True, timeout=300, env=env, cwd=ROOT)
    # dmabuf MR mode: must SUCCEED end to end on real VRAM
    assert out.returncode == 0, out.stdout + out.stderr
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    r = json.loads(line)
    assert r["config"]["transport"] == "verbs"
    assert r["config"]["verbs_mr"] == "dmabuf"
    assert r["config"]["integrity"] == "ok"


def test_two_process_fabric_into_real_vram(built):
    """Strongest no-HCA approximation of the product: TWO real
    processes; the server's region is REAL HBM exported as a dmabuf
    (its fake core mmaps the BAR window); the client process
    RDMA-writes over TCP + the shm fabric, the server's NIC-role
    engine applies the writes through the BAR mapping into VRAM, and
    the on-GPU kernels verify remotely."""
    import re
    import time

    if not os.path.exists(FAKEVERBS):
        pytest.skip("fakeverbs harness not built")
    env = dict(os.environ, FAKE_VERBS_SHM=f"/rocnr_gfab_{os.getpid()}")
    srv = subprocess.Popen(
        [FAKEVERBS, "--serve", "0", "--mr", "dmabuf",
         "--region", "67108864"],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True,
        env=env)
    try:
        port = None
        t0 = time.time()
        while time.time() - t0 < 60 and port is None:
            line = srv.stdout.readline()
            m = re.search(r"listening on port (\d+)", line)
            if m:
                port = int(m.group(1))
        assert port, "server never announced"
        out = subprocess.run(
            [FAKEVERBS, "--connect", f"127.0.0.1:{port}", "--msg",
             "1048576", "--region", "67108864", "--secs", "0.2"],
            capture_output=True, text=True, timeout=180, env=env)
        assert out.returncode == 0, out.stdout + out.stderr
        r = json.loads(out.stdout.strip().splitlines()[-1])
        assert r["remote_integrity"] == "ok"
        srv.wait(timeout=60)
        assert srv.returncode == 0
    finally:
        if srv.poll() is None:
            srv.kill()
        srv.communicate()
        shm = "/dev/shm" + env["FAKE_VERBS_SHM"]
        if os.path.exists(shm):
            os.unlink(shm)
