#!/usr/bin/env python3
"""Run the BASELINE.json config matrix end to end and emit one JSON
report — the operational closure of the measurement plan.

  1. host loopback (no GPU, no HCA)        -> fake transport
  2. 1 GPU, 1 GiB hipMalloc region, write  -> sdma (or verbs on HCA)
  3. 1 GPU sweep 4 KiB..64 MiB, both dirs, CRC-verified
  4. 2 GPUs, one QP each, aggregate        -> skipped if <2 GPUs
  5. N GPUs, 64 GiB pinned region each     -> skipped if <N GPUs or --quick

Usage: python tools/run_matrix.py [--out report.json] [--quick]
"""
from __future__ import annotations

import argparse
import json
import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="")
    ap.add_argument("--quick", action="store_true",
                    help="shrink regions/durations for smoke use")
    args = ap.parse_args()

    import torch

    from rocnrdma_amd.harness.sweep import run_point, run_sweep
    from rocnrdma_amd.transport import get_transport

    ngpu = torch.cuda.device_count() if torch.cuda.is_available() else 0
    scale = 16 if args.quick else 1
    report = {"gpus_visible": ngpu, "configs": {}}

    # config 1: host loopback
    tp = get_transport("fake", msg_bytes=1 << 20,
                       region_bytes=(256 << 20) // scale)
    r = run_point(tp, target_secs=0.3)
    r["integrity_bad"] = tp.integrity_check(seed=1)
    tp.close()
    report["configs"]["1_host_loopback"] = r

    if ngpu >= 1:
        # config 2: 1 GiB region, 64 MiB messages, write
        tp = get_transport("sdma", msg_bytes=64 << 20,
                           region_bytes=(1 << 30) // scale)
        r = run_point(tp, target_secs=0.5)
        r["integrity_bad"] = tp.integrity_check(seed=2)
        tp.close()
        report["configs"]["2_1gpu_1gib_write"] = r

        # config 3: sweep, both directions, integrity via CRC kernels
        rows = run_sweep("sdma", region_bytes=(1 << 30) // scale,
                         target_secs=0.2 if args.quick else 0.5)
        report["configs"]["3_sweep"] = rows
        # CRC cross-check on a filled region
        import rocnrdma_amd.ops as ops
        from rocnrdma_amd.utils import pattern

        buf = torch.empty((64 << 20) // scale, dtype=torch.uint8,
                          device="cuda:0")
        ops.fill_(buf, seed=3)
        crc_gpu = ops.crc32_pages(buf).cpu().numpy()
        crc_ref = pattern.crc32_pages_reference(
            pattern.fill_reference(buf.numel(), 3))
        report["configs"]["3_crc_check"] = {
            "pages": int(len(crc_gpu)),
            "match": bool((crc_gpu.view("u4") == crc_ref).all()),
        }

    if ngpu >= 2:
        # config 4: one QP per GPU via the native harness fan-out
        import subprocess

        out = subprocess.run(
            [os.path.join(ROOT, "harness", "build", "rocp2p_bw"),
             "--transport", "hip", "--msg", "67108864", "--region",
             str((1 << 30) // scale), "--secs", "0.5", "--gpus", "2",
             "--json"],
            capture_output=True, text=True, timeout=300)
        report["configs"]["4_2gpu_aggregate"] = (
            json.loads(out.stdout.strip()) if out.returncode == 0
            else {"error": out.stderr[-500:]})
    else:
        report["configs"]["4_2gpu_aggregate"] = "skipped: <2 GPUs"

    if ngpu >= 1 and not args.quick:
        # config 5 (per-GPU slice): 64 GiB region on this GPU
        tp = get_transport("sdma", msg_bytes=64 << 20,
                           region_bytes=64 << 30)
        r = run_point(tp, target_secs=1.0)
        r["integrity_bad"] = tp.integrity_check(seed=5)
        tp.close()
        report["configs"]["5_64gib_region"] = r
    else:
        report["configs"]["5_64gib_region"] = "skipped (quick or no GPU)"

    text = json.dumps(report, indent=1)
    print(text)
    if args.out:
        with open(args.out, "w") as f:
            f.write(text + "\n")
    bad = 0
    for v in report["configs"].values():
        if isinstance(v, dict) and v.get("integrity_bad"):
            bad += 1
    raise SystemExit(1 if bad else 0)


if __name__ == "__main__":
    main()
