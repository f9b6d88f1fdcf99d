"""GPU tests of the bench.py driver contract — single rank and a
2-process gloo rendezvous sharing one GPU (the multi-GPU launch shape
on a 1-GPU box)."""
import json
import os
import subprocess
import sys

import pytest

torch = pytest.importorskip("torch")

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU"),
]

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(ROOT, "bench.py")

SMALL = ["--msg-bytes", "4194304", "--region-bytes", "67108864",
         "--steps", "3", "--warmup", "1"]


def _env(extra=None):
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    if extra:
        env.update(extra)
    return env


@pytest.mark.timeout(600)
def test_bench_single_rank_gpu():
    out = subprocess.run([sys.executable, BENCH] + SMALL,
                         capture_output=True, text=True, env=_env(),
                         cwd=ROOT, timeout=480)
    assert out.returncode == 0, out.stderr
    r = json.loads([l for l in out.stdout.splitlines()
                    if l.startswith("{")][0])
    assert r["config"]["transport"] == "sdma"
    assert r["config"]["integrity"] == "ok"
    assert r["value"] > 5  # GB/s; PCIe path


@pytest.mark.timeout(600)
def test_bench_two_ranks_one_gpu():
    """Two ranks rendezvous over gloo at 127.0.0.1, both on cuda:0 —
    validates the distributed launch path the driver uses for N>1."""
    procs = []
    for rank in range(2):
        env = _env({"RANK": str(rank), "WORLD_SIZE": "2",
                    "LOCAL_RANK": "0", "MASTER_ADDR": "127.0.0.1",
                    "MASTER_PORT": "29612"})
        procs.append(subprocess.Popen(
            [sys.executable, BENCH, "--gpus", "2"] + SMALL,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True,
            env=env, cwd=ROOT))
    outs = [p.communicate(timeout=480) for p in procs]
    for p, (so, se) in zip(procs, outs):
        assert p.returncode == 0, se
    r = json.loads([l for l in outs[0][0].splitlines()
                    if l.startswith("{")][0])
    assert r["n_gpus"] == 2
    assert r["config"]["integrity"] == "ok"
