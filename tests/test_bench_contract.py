"""bench.py contract tests (CPU, fake transport): single-rank JSON line
and 2-process gloo aggregate."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(ROOT, "bench.py")

REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


def run_bench(extra, env_extra=None, timeout=240):
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    if env_extra:
        env.update(env_extra)
    out = subprocess.run(
        [sys.executable, BENCH, "--transport", "fake", "--msg-bytes",
         "65536", "--region-bytes", "1048576", "--steps", "3",
         "--warmup", "1"] + extra,
        capture_output=True, text=True, env=env, cwd=ROOT, timeout=timeout)
    assert out.returncode == 0, out.stderr
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    return json.loads(lines[0])


def test_single_rank_json_contract():
    r = run_bench([])
    for key in REQUIRED:
        assert key in r, key
    assert r["unit"] == "GB/s"
    assert r["value"] > 0
    assert r["steps"] == 3 and r["warmup"] == 1
    assert r["higher_is_better"] is True
    assert r["scaling"] == "weak"
    assert r["vs_baseline"] is None
    assert r["data"] == "synthetic"
    assert r["config"]["transport"] == "fake"
    assert r["config"]["integrity"] == "ok"
    assert r["ms_per_step"] > 0
    assert "ib_write_bw" in r["metric"]


def test_integrity_gate():
    # direction=read also exercises the pull path
    r = run_bench(["--direction", "read"])
    assert r["config"]["integrity"] == "ok"


@pytest.mark.timeout(300)
def test_two_rank_gloo_aggregate():
    """Two fake-transport ranks over gloo: value aggregates both ranks."""
    procs = []
    port = "29533"
    for rank in range(2):
        env = dict(os.environ)
        env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
        env.update({"RANK": str(rank), "WORLD_SIZE": "2",
                    "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
                    "MASTER_PORT": port})
        procs.append(subprocess.Popen(
            [sys.executable, BENCH, "--transport", "fake", "--gpus", "2",
             "--msg-bytes", "65536", "--region-bytes", "1048576",
             "--steps", "3", "--warmup", "1"],
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True,
            env=env, cwd=ROOT))
    outs = [p.communicate(timeout=240) for p in procs]
    for p, (so, se) in zip(procs, outs):
        assert p.returncode == 0, se
    # rank 0 prints the line
    lines = [l for l in outs[0][0].splitlines() if l.startswith("{")]
    assert len(lines) == 1
    r = json.loads(lines[0])
    assert r["n_gpus"] == 2
    assert "x2" in r["config"]["parallelism"]


def test_msgs_per_step_override():
    r = run_bench(["--msgs-per-step", "4"])
    assert r["config"]["global_batch"] == 4
    assert r["value"] > 0


def test_msg_sweep_embedded_single_rank():
    r = run_bench([])
    sw = r["config"]["msg_sweep_gbps"]
    assert sw is not None
    assert str(r["config"]["msg_bytes"]) in sw
    assert "4096" in sw and all(v > 0 for v in sw.values())


FAKEVERBS_BIN = os.path.join(ROOT, "harness", "build",
                             "rocp2p_bw_fakeverbs")


def test_auto_transport_order():
    """--transport auto must prefer the real verbs data
    plane whenever a usable stack exists — never hard-pick sdma on a
    GPU box."""
    sys.path.insert(0, ROOT)
    import bench

    assert bench.choose_transport(True, True) == "verbs"
    assert bench.choose_transport(False, True) == "verbs"
    assert bench.choose_transport(True, False) == "sdma"
    assert bench.choose_transport(False, False) == "fake"


@pytest.mark.timeout(300)
def test_auto_selects_verbs_and_reports_it():
    """With a verbs stack present (fake-verbs CI layer + forced
    availability), `--transport auto` must route through the native
    verbs data plane and emit transport=verbs in the JSON contract."""
    if not os.path.exists(FAKEVERBS_BIN):
        pytest.skip("fakeverbs harness not built")
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    env["ROCNR_FORCE_VERBS"] = "1"
    env["ROCNR_VERBS_HARNESS"] = FAKEVERBS_BIN
    out = subprocess.run(
        [sys.executable, BENCH, "--gpus", "1", "--steps", "3",
         "--warmup", "1", "--msg-bytes", "1048576",
         "--region-bytes", "16777216"],
        capture_output=True, text=True, timeout=240, env=env, cwd=ROOT)
    assert out.returncode == 0, out.stderr
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    r = json.loads(line)
    assert r["config"]["transport"] == "verbs"
    assert r["config"]["integrity"] == "ok"
    assert r["value"] > 0
    assert r["steps"] == 3 and r["warmup"] == 1
    # the metric's three sizes are present for single-rank runs
    assert set(r["config"]["msg_sweep_gbps"]) == {"4096", "1048576"}


@pytest.mark.timeout(300)
def test_two_rank_verbs_aggregate():
    """Two verbs ranks (fake-verbs layer) over gloo: the whole-job
    aggregate covers both ranks' native runs, max-elapsed over ranks."""
    if not os.path.exists(FAKEVERBS_BIN):
        pytest.skip("fakeverbs harness not built")
    procs = []
    port = "29537"
    for rank in range(2):
        env = dict(os.environ)
        env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
        env.update({"RANK": str(rank), "WORLD_SIZE": "2",
                    "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
                    "MASTER_PORT": port, "ROCNR_FORCE_VERBS": "1",
                    "ROCNR_VERBS_HARNESS": FAKEVERBS_BIN})
        procs.append(subprocess.Popen(
            [sys.executable, BENCH, "--gpus", "2", "--msg-bytes", "65536",
             "--region-bytes", "1048576", "--steps", "3", "--warmup", "1"],
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True,
            env=env, cwd=ROOT))
    outs = [p.communicate(timeout=240) for p in procs]
    for p, (so, se) in zip(procs, outs):
        assert p.returncode == 0, se
    lines = [l for l in outs[0][0].splitlines() if l.startswith("{")]
    assert len(lines) == 1
    r = json.loads(lines[0])
    assert r["config"]["transport"] == "verbs"
    assert r["n_gpus"] == 2
    assert "x2" in r["config"]["parallelism"]


@pytest.mark.timeout(300)
def test_eight_rank_gloo_aggregate():
    """8-rank fan-out math (the 8-GPU node shape) on
    CPU — whole-job value is sum over ranks of bytes / max elapsed,
    parallelism records x8."""
    procs = []
    port = "29541"
    for rank in range(8):
        env = dict(os.environ)
        env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
        env.update({"RANK": str(rank), "WORLD_SIZE": "8",
                    "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
                    "MASTER_PORT": port})
        procs.append(subprocess.Popen(
            [sys.executable, BENCH, "--transport", "fake", "--gpus", "8",
             "--msg-bytes", "16384", "--region-bytes", "262144",
             "--steps", "2", "--warmup", "1"],
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True,
            env=env, cwd=ROOT))
    outs = [p.communicate(timeout=240) for p in procs]
    for p, (so, se) in zip(procs, outs):
        assert p.returncode == 0, se
    lines = [l for l in outs[0][0].splitlines() if l.startswith("{")]
    assert len(lines) == 1
    r = json.loads(lines[0])
    assert r["n_gpus"] == 8
    assert r["config"]["parallelism"] == "1qp-per-gpu x8"
    # aggregate must be the whole-job sum: 8 ranks x 2 steps x 16 msgs
    # x 16 KiB over the max elapsed -> global_batch = msgs/step x 8
    assert r["config"]["global_batch"] == 16 * 8
    assert r["value"] > 0


@pytest.mark.timeout(300)
def test_verbs_read_direction():
    """bench --direction read routes through the native verbs plane
    (RDMA READ WRs) and keeps the JSON contract."""
    if not os.path.exists(FAKEVERBS_BIN):
        pytest.skip("fakeverbs harness not built")
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    env["ROCNR_FORCE_VERBS"] = "1"
    env["ROCNR_VERBS_HARNESS"] = FAKEVERBS_BIN
    out = subprocess.run(
        [sys.executable, BENCH, "--gpus", "1", "--steps", "2",
         "--warmup", "1", "--msg-bytes", "1048576",
         "--region-bytes", "16777216", "--direction", "read"],
        capture_output=True, text=True, timeout=240, env=env, cwd=ROOT)
    assert out.returncode == 0, out.stderr
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    r = json.loads(line)
    assert r["config"]["transport"] == "verbs"
    assert r["config"]["direction"] == "read"
    assert r["config"]["integrity"] == "ok"
