"""Two-process client/server bandwidth run (ib_write_bw server/client
shape).  Same-host data plane is POSIX shm; the OOB exchange is the
one a remote verbs deployment uses.

  server:  python -m rocnrdma_amd.harness.remote --serve --region 268435456
  client:  python -m rocnrdma_amd.harness.remote --connect 127.0.0.1:PORT \
               --msg 65536 --region 268435456 --secs 1
"""
from __future__ import annotations

import argparse

from rocnrdma_amd.harness.sweep import run_point
from rocnrdma_amd.transport.shm import ShmInitiatorTransport, target_serve


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--serve", action="store_true")
    ap.add_argument("--connect", default="")
    ap.add_argument("--port", type=int, default=0)
    ap.add_argument("--region", type=int, default=256 << 20)
    ap.add_argument("--msg", type=int, default=64 << 10)
    ap.add_argument("--secs", type=float, default=1.0)
    ap.add_argument("--direction", default="write")
    args = ap.parse_args()

    if args.serve:
        target_serve(None, args.region, port=args.port)
        return
    if not args.connect:
        ap.error("need --serve or --connect HOST:PORT")
    host, port = args.connect.rsplit(":", 1)
    tp = ShmInitiatorTransport(msg_bytes=args.msg,
                               region_bytes=args.region, host=host,
                               port=int(port), direction=args.direction)
    r = run_point(tp, target_secs=args.secs)
    bad = tp.integrity_check(seed=0xABCD)
    r["remote_verify_bad"] = bad
    tp.close()
    print(r)
    raise SystemExit(1 if bad else 0)


if __name__ == "__main__":
    main()
