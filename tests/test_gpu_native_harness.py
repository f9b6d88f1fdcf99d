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
