"""PCIe/NUMA locality helpers for multi-GPU fan-out.

The reference's operational guidance — keep the HCA and GPU on one root
complex (reference README.md:71-72) — generalizes on an 8×MI355X node
to: pin each rank's CPU threads to the NUMA node of its GPU, so pinned
staging, descriptor rings and the posting loop live on the memory/PCIe
complex the DMA traffic crosses.  Best-effort: every probe degrades to
no-op off-Linux or without sysfs.
"""
from __future__ import annotations

import os
import re


def parse_cpulist(text: str) -> list[int]:
    """Parse a sysfs cpulist ("0-3,8,10-11") into CPU ids."""
    cpus: list[int] = []
    for part in text.strip().split(","):
        if not part:
            continue
        if "-" in part:
            lo, hi = part.split("-")
            cpus.extend(range(int(lo), int(hi) + 1))
        else:
            cpus.append(int(part))
    return cpus


def gpu_pci_bus_id(device_index: int) -> str | None:
    """PCI BDF of a torch CUDA/HIP device, if discoverable."""
    try:
        import torch

        props = torch.cuda.get_device_properties(device_index)
        bus = getattr(props, "pci_bus_id", None)
        dom = getattr(props, "pci_domain_id", 0) or 0
        dev = getattr(props, "pci_device_id", None)
        if bus is None or dev is None:
            return None
        return f"{dom:04x}:{bus:02x}:{dev:02x}.0"
    except Exception:
        return None


def numa_node_of_pci(bdf: str) -> int | None:
    path = f"/sys/bus/pci/devices/{bdf}/numa_node"
    try:
        with open(path) as f:
            node = int(f.read().strip())
        return node if node >= 0 else None
    except OSError:
        return None


def cpus_of_numa_node(node: int) -> list[int]:
    try:
        with open(f"/sys/devices/system/node/node{node}/cpulist") as f:
            return parse_cpulist(f.read())
    except OSError:
        return []


def bind_rank_near_gpu(device_index: int) -> int | None:
    """Pin this process to the CPUs of its GPU's NUMA node.

    Returns the node bound to, or None if topology was not
    discoverable (no-op in that case).
    """
    bdf = gpu_pci_bus_id(device_index)
    if not bdf:
        return None
    node = numa_node_of_pci(bdf)
    if node is None:
        return None
    cpus = cpus_of_numa_node(node)
    if not cpus:
        return None
    try:
        os.sched_setaffinity(0, cpus)
    except (AttributeError, OSError):
        return None
    return node


def describe() -> list[dict]:
    """Topology summary: one row per visible GPU."""
    rows = []
    try:
        import torch

        n = torch.cuda.device_count() if torch.cuda.is_available() else 0
    except ImportError:
        n = 0
    for i in range(n):
        bdf = gpu_pci_bus_id(i)
        rows.append({
            "gpu": i,
            "pci": bdf,
            "numa": numa_node_of_pci(bdf) if bdf else None,
        })
    return rows
