"""Message-size bandwidth sweep — the ib_write_bw/ib_read_bw-style
matrix (BASELINE configs 3): 4 KB … 64 MB x {write, read}, one QP
analog per GPU.

CLI:  python -m rocnrdma_amd.harness.sweep [--transport auto] [--out f.json]
"""
from __future__ import annotations

import argparse
import json
import time

DEFAULT_SIZES = [4 << 10, 64 << 10, 1 << 20, 16 << 20, 64 << 20]


def run_point(tp, target_secs: float = 1.0, min_msgs: int = 8) -> dict:
    """Time-bound bandwidth measurement of one transport config."""
    import torch

    has_gpu = torch.cuda.is_available()

    def sync():
        tp.flush()
        if has_gpu:
            torch.cuda.synchronize()

    # warmup: one region pass (capped)
    tp.post_many(0, min(tp.msgs_per_region, max(64, tp.inflight)))
    sync()

    # calibrate burst so each timed burst is >= ~50 ms
    burst = max(min_msgs, tp.inflight)
    posted = 0
    t_end = time.perf_counter() + target_secs
    t0 = time.perf_counter()
    while time.perf_counter() < t_end:
        tp.post_many(posted, burst)
        posted += burst
        tp.flush()
    sync()
    elapsed = time.perf_counter() - t0
    gbps = posted * tp.msg_bytes / elapsed / 1e9
    return {
        "msg_bytes": tp.msg_bytes,
        "direction": tp.direction,
        "msgs": posted,
        "secs": round(elapsed, 4),
        "gbps": round(gbps, 3),
        "mops": round(posted / elapsed / 1e6, 4),
    }


def run_lat_point(tp, iters: int = 1000) -> dict:
    """ib_write_lat analog: single-message completion latency."""
    import time as _t

    tp.post_many(0, 1)
    tp.flush()
    us = []
    for i in range(iters):
        t0 = _t.perf_counter()
        tp.post_many(i, 1)
        tp.flush()
        us.append((_t.perf_counter() - t0) * 1e6)
    us.sort()
    return {
        "msg_bytes": tp.msg_bytes,
        "direction": tp.direction,
        "mode": "lat",
        "iters": iters,
        "us_min": round(us[0], 2),
        "us_p50": round(us[len(us) // 2], 2),
        "us_p99": round(us[int(len(us) * 0.99)], 2),
        "us_max": round(us[-1], 2),
    }


def run_sweep(transport: str = "auto", region_bytes: int = 1 << 30,
              sizes=None, directions=("write", "read"),
              target_secs: float = 1.0, device=None,
              num_streams: int = 2, inflight: int | None = 0,
              lat_iters: int = 0) -> list[dict]:
    from rocnrdma_amd.transport import get_transport

    rows = []
    for direction in directions:
        for msg in sizes or DEFAULT_SIZES:
            region = max(region_bytes // msg, 1) * msg
            tp = get_transport(transport, msg_bytes=msg,
                               region_bytes=region, direction=direction,
                               device=device, num_streams=num_streams,
                               inflight=inflight)
            try:
                if lat_iters:
                    row = run_lat_point(tp, lat_iters)
                else:
                    row = run_point(tp, target_secs=target_secs)
                row["transport"] = tp.name
                rows.append(row)
            finally:
                tp.close()
    return rows


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--transport", default="auto")
    ap.add_argument("--region-bytes", type=int, default=1 << 30)
    ap.add_argument("--secs", type=float, default=1.0)
    ap.add_argument("--streams", type=int, default=2)
    ap.add_argument("--inflight", type=int, default=0)
    ap.add_argument("--sizes", default="")
    ap.add_argument("--out", default="")
    ap.add_argument("--lat", type=int, default=0,
                    help="latency mode: N single-message iterations")
    args = ap.parse_args()

    sizes = ([int(s) for s in args.sizes.split(",")] if args.sizes
             else DEFAULT_SIZES)
    import torch

    if args.transport in ("auto", "fake") and not torch.cuda.is_available():
        args.region_bytes = min(args.region_bytes, 256 << 20)
    rows = run_sweep(args.transport, args.region_bytes, sizes,
                     target_secs=args.secs, num_streams=args.streams,
                     inflight=args.inflight, lat_iters=args.lat)
    if args.lat:
        print(f"{'msg':>12} {'dir':>6} {'us_min':>8} {'us_p50':>8} "
              f"{'us_p99':>8} {'us_max':>8}")
        for r in rows:
            print(f"{r['msg_bytes']:>12} {r['direction']:>6} "
                  f"{r['us_min']:>8.2f} {r['us_p50']:>8.2f} "
                  f"{r['us_p99']:>8.2f} {r['us_max']:>8.2f}")
    else:
        print(f"{'msg':>12} {'dir':>6} {'GB/s':>10} {'Mmsg/s':>10}")
        for r in rows:
            print(f"{r['msg_bytes']:>12} {r['direction']:>6} "
                  f"{r['gbps']:>10.3f} {r['mops']:>10.4f}")
    if args.out:
        with open(args.out, "w") as f:
            json.dump(rows, f, indent=1)


if __name__ == "__main__":
    main()
