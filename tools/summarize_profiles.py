#!/usr/bin/env python3
"""One-screen summary of the committed measurement artifacts
(profiles/*.json) — the quick 'what did we measure' view.

Usage: python tools/summarize_profiles.py
"""
from __future__ import annotations

import glob
import json
import os

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
P = os.path.join(ROOT, "profiles")


def load(name):
    path = os.path.join(P, name)
    if not os.path.exists(path):
        return None
    with open(path) as f:
        return json.load(f)


def main():
    b = load("bench_n1.json")
    if b:
        print(f"headline bench      : {b['value']} {b['unit']} "
              f"({b['config']['msg_bytes']} B msgs, "
              f"integrity {b['config']['integrity']})")
        sweep = b["config"].get("msg_sweep_gbps")
        if sweep:
            pts = ", ".join(f"{int(k) >> 10} KiB={v}" for k, v in
                            sorted(sweep.items(), key=lambda kv: int(kv[0])))
            print(f"  msg points        : {pts}")
    s = load("sweep_1gpu.json")
    if s:
        for row in s:
            if row["direction"] == "write":
                print(f"sweep write {row['msg_bytes']:>9} B : "
                      f"{row['gbps']:7.2f} GB/s  ({row['mops']:.3f} Mmsg/s)")
    d = load("devbw_2gib.json")
    if d:
        print("device kernels (2 GiB): " + ", ".join(
            f"{k.split('_')[0]}={v:.0f}" for k, v in d.items()
            if k.endswith("GBps") or k.endswith("GBps_rw")))
    d = load("devbw_240gib.json")
    if d:
        print("device kernels (240 GiB footprint): " + ", ".join(
            f"{k.split('_')[0]}={v:.0f}" for k, v in d.items()
            if "GBps" in k))
    for f in sorted(glob.glob(os.path.join(P, "soak_*.txt"))):
        with open(f) as fh:
            print(os.path.basename(f), "->", fh.read().strip()[:100])
    m = load("config_matrix_1gpu.json")
    if m:
        done = [k for k, v in m["configs"].items()
                if not isinstance(v, str)]
        print(f"config matrix       : {len(done)} configs measured "
              f"({', '.join(done)})")

    # --- round 2 ---
    b = load("r2_bench_64g_region.json")
    if b:
        print(f"r2 64 GiB region    : {b['value']} GB/s "
              f"(integrity {b['config']['integrity']})")
    b = load("bench_sustained_r2.json")
    if b:
        tb = b['steps'] * 1.074 / 1000  # 1 GiB region per step
        print(f"r2 5-min sustained  : {b['value']} GB/s "
              f"({b['steps']} steps, ~{tb:.1f} TB)")
    b = load("bench_verbs_2rank.json")
    if b:
        print(f"r2 bench verbs path : {b['value']} GB/s aggregate, "
              f"{b['config']['parallelism']}, mr="
              f"{b['config'].get('verbs_mr')} (real VRAM dmabuf)")
    for name, label in (("dmabuf_write.json", "r2 dmabuf BAR write"),
                        ("dmabuf_16g.json", "r2 dmabuf 16 GiB")):
        d = load(name)
        if d:
            print(f"{label:<20}: {d['gbps']} GB/s "
                  f"(integrity {d['integrity']})")
    for f in sorted(glob.glob(os.path.join(P, "r2_soak_*.txt"))):
        with open(f) as fh:
            print(os.path.basename(f), "->", fh.read().strip()[-110:])


if __name__ == "__main__":
    main()
