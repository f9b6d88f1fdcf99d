"""Soak / stress: randomized message sizes, directions and burst
shapes against one region, with periodic full CRC audits — the
stability tier for production deployment (run for minutes/hours on a
deploy candidate; the GPU test tier runs a short bounded pass).

CLI: python -m rocnrdma_amd.harness.soak [--secs 30] [--transport auto]
Exit nonzero on any integrity failure.
"""
from __future__ import annotations

import argparse
import random
import time

SIZES = [4 << 10, 16 << 10, 64 << 10, 256 << 10, 1 << 20, 4 << 20]


def run_soak(transport: str = "auto", secs: float = 10.0,
             region_bytes: int = 256 << 20, seed: int = 1234,
             device=None, metrics=None) -> dict:
    from rocnrdma_amd.transport import get_transport

    if metrics is None:
        from rocnrdma_amd.utils.metrics import TransferMetrics

        metrics = TransferMetrics(port=None)
    rng = random.Random(seed)
    t_end = time.monotonic() + secs
    stats = {"cycles": 0, "msgs": 0, "bytes": 0, "audits": 0,
             "failures": 0}
    while time.monotonic() < t_end:
        msg = rng.choice(SIZES)
        direction = rng.choice(["write", "read"])
        region = max(region_bytes // msg, 1) * msg
        tp = get_transport(transport, msg_bytes=msg, region_bytes=region,
                           direction=direction, device=device)
        try:
            # a few random bursts (bandwidth phase, content irrelevant)
            posted = 0
            for _ in range(rng.randint(1, 4)):
                burst = rng.randint(1, max(2, tp.inflight))
                tp.post_many(posted, burst)
                posted += burst
                tp.flush()
            # audit: full-region pattern transfer + receiver verify
            bad = tp.integrity_check(seed=rng.getrandbits(32))
            stats["audits"] += 1
            if bad:
                stats["failures"] += 1
            moved = (posted + tp.msgs_per_region) * msg
            stats["msgs"] += posted + tp.msgs_per_region
            stats["bytes"] += moved
            stats["cycles"] += 1
            metrics.observe_bytes(direction, moved,
                                  posted + tp.msgs_per_region)
            metrics.observe_audit(ok=bad == 0)
        finally:
            tp.close()
    return stats


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--secs", type=float, default=30.0)
    ap.add_argument("--transport", default="auto")
    ap.add_argument("--region-bytes", type=int, default=256 << 20)
    ap.add_argument("--seed", type=int, default=1234)
    ap.add_argument("--metrics-port", type=int, default=0,
                    help="expose Prometheus metrics on this port")
    args = ap.parse_args()
    from rocnrdma_amd.utils.metrics import TransferMetrics

    metrics = TransferMetrics(port=args.metrics_port or None)
    stats = run_soak(args.transport, args.secs, args.region_bytes,
                     args.seed, metrics=metrics)
    print(stats)
    raise SystemExit(1 if stats["failures"] else 0)


if __name__ == "__main__":
    main()
