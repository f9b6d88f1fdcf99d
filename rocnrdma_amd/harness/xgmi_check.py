"""Intra-node fabric cross-check (xGMI matters to this
project only as the reason aggregate multi-GPU NIC bandwidth isn't
staging-bound; an RCCL sanity config cross-checks the intra-node
ceiling).

Two probes:
 - p2p:  hipMemcpyPeer-style device-to-device copy bandwidth between
         every GPU pair visible to this process (xGMI links);
 - rccl: all_reduce bus bandwidth via torch.distributed (backend
         "nccl" == RCCL on ROCm) when launched with world_size > 1.

CLI: python -m rocnrdma_amd.harness.xgmi_check [--mb 256]
(single-GPU boxes report the self-copy ceiling only)
"""
from __future__ import annotations

import argparse
import json
import os
import time


def p2p_matrix(mb: int = 256, iters: int = 5) -> list[dict]:
    import torch

    n = torch.cuda.device_count()
    nbytes = mb << 20
    rows = []
    bufs = []
    for d in range(n):
        with torch.cuda.device(d):
            bufs.append(torch.empty(nbytes, dtype=torch.uint8,
                                    device=f"cuda:{d}"))
    for src in range(n):
        for dst in range(n):
            with torch.cuda.device(dst):
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                for _ in range(iters):
                    bufs[dst].copy_(bufs[src], non_blocking=True)
                torch.cuda.synchronize()
                dt = time.perf_counter() - t0
            rows.append({"src": src, "dst": dst,
                         "gbps": round(nbytes * iters / dt / 1e9, 1)})
    return rows


def rccl_allreduce_bw(mb: int = 256, iters: int = 10) -> dict | None:
    """Requires torchrun launch (world>1); returns None otherwise."""
    import torch

    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world < 2 or not torch.cuda.is_available():
        return None
    import torch.distributed as dist

    rank = int(os.environ.get("RANK", "0"))
    local = int(os.environ.get("LOCAL_RANK", str(rank)))
    torch.cuda.set_device(local)
    dist.init_process_group("nccl")
    x = torch.ones(mb << 18, dtype=torch.float32, device="cuda")  # mb MiB
    for _ in range(3):
        dist.all_reduce(x)
    torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        dist.all_reduce(x)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    nbytes = x.numel() * 4
    # ring all_reduce bus bytes: 2*(n-1)/n per rank
    bus = 2 * (world - 1) / world * nbytes * iters / dt
    dist.destroy_process_group()
    return {"world": world, "algbw_GBps": round(nbytes * iters / dt / 1e9, 1),
            "busbw_GBps": round(bus / 1e9, 1)}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mb", type=int, default=256)
    args = ap.parse_args()
    import torch

    out = {}
    if torch.cuda.is_available():
        out["p2p"] = p2p_matrix(args.mb)
    r = rccl_allreduce_bw(args.mb)
    if r:
        out["rccl_allreduce"] = r
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
