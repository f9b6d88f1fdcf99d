"""Tests for the repo tooling: rocpd profile summarizer and the config
matrix runner (quick mode)."""
import os
import sqlite3
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(ROOT, "tools"))


def make_fake_rocpd(path):
    sfx = "deadbeef"
    db = sqlite3.connect(path)
    db.execute(f"CREATE TABLE rocpd_kernel_dispatch_{sfx} "
               "(id INTEGER, kernel_id INTEGER, start INTEGER, end INTEGER)")
    db.execute(f"CREATE TABLE rocpd_info_kernel_symbol_{sfx} "
               "(id INTEGER, display_name TEXT)")
    db.execute(f"CREATE TABLE rocpd_string_{sfx} (id INTEGER, string TEXT)")
    db.execute(f"CREATE TABLE rocpd_memory_copy_{sfx} "
               "(name_id INTEGER, start INTEGER, end INTEGER, size INTEGER)")
    db.execute(f"INSERT INTO rocpd_info_kernel_symbol_{sfx} VALUES "
               "(1, 'k_fill(ulong*, ulong, ulong)')")
    for i, (s, e) in enumerate([(0, 1000), (2000, 3500), (5000, 5500)]):
        db.execute(f"INSERT INTO rocpd_kernel_dispatch_{sfx} VALUES "
                   f"({i}, 1, {s}, {e})")
    db.execute(f"INSERT INTO rocpd_string_{sfx} VALUES (9, "
               "'MEMORY_COPY_HOST_TO_DEVICE')")
    db.execute(f"INSERT INTO rocpd_memory_copy_{sfx} VALUES "
               "(9, 0, 1000000, 1000000)")
    db.commit()
    db.close()


def test_rocpd_stats_summarizes(tmp_path):
    import rocpd_stats

    db = tmp_path / "r.db"
    make_fake_rocpd(str(db))
    rows = rocpd_stats.kernel_stats(str(db))
    assert len(rows) == 1
    assert rows[0]["name"] == "k_fill"
    assert rows[0]["calls"] == 3
    assert rows[0]["total_ms"] == 0.003
    mc = rocpd_stats.memcpy_stats(str(db))
    assert mc[0]["op"] == "MEMORY_COPY_HOST_TO_DEVICE"
    assert mc[0]["GBps"] == 1.0


@pytest.mark.timeout(300)
def test_run_matrix_quick_cpu():
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "tools", "run_matrix.py"),
         "--quick"],
        capture_output=True, text=True, cwd=ROOT, timeout=240,
        env=dict(os.environ,
                 PYTHONPATH=ROOT + os.pathsep +
                 os.environ.get("PYTHONPATH", "")))
    assert out.returncode == 0, out.stderr
    import json

    r = json.loads(out.stdout)
    assert r["configs"]["1_host_loopback"]["integrity_bad"] == 0


MODERN_PEER = """
struct peer_memory_client {
  int (*get_pages)(unsigned long addr, size_t size, int write, int force,
                   struct sg_table *sg_head, void *client_context,
                   u64 core_context);
};
"""
LEGACY_PEER = """
struct peer_memory_client {
  int (*get_pages)(unsigned long addr, size_t size, int write, int force,
                   struct sg_table *sg_head, void *client_context,
                   void *core_context);
};
"""
MODERN_AMDR = """
struct amd_rdma_interface {
  int (*get_pages)(uint64_t address, uint64_t length, struct pid *pid,
                   struct device *dma_dev,
                   struct amd_p2p_info **amd_p2p_data,
                   void (*free_callback)(void *), void *client_priv);
};
"""
LEGACY_AMDR = """
struct amd_rdma_interface {
  int (*get_pages)(uint64_t address, uint64_t length, struct pid *pid,
                   struct amd_p2p_info **amd_p2p_data,
                   void (*free_callback)(void *), void *client_priv);
};
"""


@pytest.mark.parametrize("peer,amdr,u64,dmadev", [
    (MODERN_PEER, MODERN_AMDR, 1, 1),
    (LEGACY_PEER, LEGACY_AMDR, 0, 0),
    (MODERN_PEER, LEGACY_AMDR, 1, 0),
])
def test_abi_probe_classifies(tmp_path, peer, amdr, u64, dmadev):
    p = tmp_path / "peer_mem.h"
    a = tmp_path / "amd_rdma.h"
    p.write_text(peer)
    a.write_text(amdr)
    out = subprocess.run(
        ["sh", os.path.join(ROOT, "tools", "abi_probe.sh"), str(p), str(a)],
        capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stdout + out.stderr
    assert f"ROCNR_PEER_MEM_CORE_CONTEXT_U64={u64}" in out.stdout
    assert f"ROCNR_AMD_RDMA_HAS_DMA_DEV={dmadev}" in out.stdout


def test_first_hca_bringup_dry_run():
    """The executable bring-up sequence (RUNBOOK as a script) must
    dry-run cleanly: artifact checks pass and every step prints."""
    import subprocess
    out = subprocess.run(
        ["bash", os.path.join(ROOT, "tools", "first_hca_bringup.sh"),
         "--dry-run"],
        capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "DONE" in out.stdout
    assert "chain sweep" in out.stdout
