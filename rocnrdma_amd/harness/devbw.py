"""Device-kernel throughput microbench: fill / verify / CRC32 / copy on
HBM3E.  Context numbers for the rocprof evidence in profiles/ (HBM
ceiling ≈6.3 TB/s measured float4 copy per MI355X_MICROARCH.md).

CLI:  python -m rocnrdma_amd.harness.devbw [--mb 1024] [--out f.json]
"""
from __future__ import annotations

import argparse
import json


def bench_kernels(nbytes: int, iters: int = 10) -> dict:
    import torch

    import rocnrdma_amd.ops as ops

    dev = torch.device("cuda", 0)
    buf = torch.empty(nbytes, dtype=torch.uint8, device=dev)
    dst = torch.empty(nbytes, dtype=torch.uint8, device=dev)

    def timeit(fn, moved_bytes):
        start, end = torch.cuda.Event(True), torch.cuda.Event(True)
        fn()  # warm
        torch.cuda.synchronize()
        start.record()
        for _ in range(iters):
            fn()
        end.record()
        torch.cuda.synchronize()
        secs = start.elapsed_time(end) / 1e3 / iters
        return round(moved_bytes / secs / 1e9, 1)

    out = {"nbytes": nbytes}
    out["fill_GBps"] = timeit(lambda: ops.fill_(buf, 1), nbytes)
    out["verify_GBps"] = timeit(lambda: ops.verify(buf, 1), nbytes)
    out["crc32_GBps"] = timeit(lambda: ops.crc32_pages(buf), nbytes)
    out["copy_GBps_rw"] = timeit(lambda: ops.copy_(dst, buf), 2 * nbytes)
    out["copy_nt_GBps_rw"] = timeit(lambda: ops.copy_nt_(dst, buf),
                                    2 * nbytes)
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mb", type=int, default=1024)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--out", default="")
    args = ap.parse_args()
    res = bench_kernels(args.mb << 20, args.iters)
    print(json.dumps(res, indent=1))
    if args.out:
        with open(args.out, "w") as f:
            json.dump(res, f, indent=1)


if __name__ == "__main__":
    main()
