"""Transports: the data path an RDMA write/read takes, behind one API.

The reference's runtime data path is HCA-driven DMA between the NIC and
GPU BAR (zero bridge code at transfer time — SURVEY.md §3.2).  On boxes
with an HCA the `verbs` transport drives real ibv RDMA; on GPU-only
boxes `sdma` measures the same PCIe BAR path using the GPU's SDMA
engines (host-pinned <-> HBM async copies); `fake` is a CPU loopback for
CI.
"""
from __future__ import annotations

from .base import Transport


def available_transports() -> list[str]:
    names = ["fake", "shm"]
    try:
        import torch

        if torch.cuda.is_available():
            names.append("sdma")
    except ImportError:
        pass
    from .verbs import verbs_available

    if verbs_available():
        names.append("verbs")
    return names


def get_transport(name: str, **kw) -> Transport:
    if name == "auto":
        # auto picks the best IN-PROCESS plane (sdma/fake).  The verbs
        # data plane lives in the native harness and is routed by
        # bench.py (choose_transport / run_verbs) — constructing
        # VerbsTransport here would raise by design.
        avail = available_transports()
        name = "sdma" if "sdma" in avail else "fake"
    if name == "fake":
        from .fake import FakeTransport

        return FakeTransport(**kw)
    if name == "sdma":
        from .sdma import SdmaTransport

        return SdmaTransport(**kw)
    if name == "verbs":
        from .verbs import VerbsTransport

        return VerbsTransport(**kw)
    if name == "shm":
        from .shm import ShmInitiatorTransport

        return ShmInitiatorTransport(**kw)
    raise ValueError(f"unknown transport {name!r}")
