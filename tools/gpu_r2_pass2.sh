#!/bin/bash
# Round-2 second GPU pass: validate the round-2 tree end to end and
# refresh rocprof evidence.  Run via gpurun from the repo root.
set -u
mkdir -p gpurun_out/r2
exec > >(tee gpurun_out/r2/pass2.log) 2>&1
cd /tmp && export TMPDIR=/tmp && cd /root/repo

echo "=== pytest -m gpu (round-2 tree) ==="
timeout 700 python -m pytest tests -m gpu -x -q 2>&1 | tail -4

echo "=== smoke() ==="
timeout 300 python __graft_entry__.py smoke 2>&1 | tail -2

echo "=== rocprof: bench sdma path (zero host-bounce refresh) ==="
(cd /tmp && timeout 420 rocprofv3 --kernel-trace --memory-copy-trace \
    -d /root/repo/gpurun_out/r2/prof_bench -o bench -- \
    python /root/repo/bench.py --steps 5 --warmup 2 \
    --msg-bytes 4194304 --region-bytes 268435456 2>&1 | tail -2)
DB=$(ls gpurun_out/r2/prof_bench/*bench*.db 2>/dev/null | head -1)
[ -n "$DB" ] && python tools/rocpd_stats.py "$DB" \
    --out gpurun_out/r2/rocprof_bench_r2.md && \
    tail -20 gpurun_out/r2/rocprof_bench_r2.md

echo "=== rocprof: dmabuf BAR write path ==="
(cd /tmp && timeout 300 rocprofv3 --kernel-trace --memory-copy-trace \
    -d /root/repo/gpurun_out/r2/prof_dmabuf -o dmabuf -- \
    /root/repo/harness/build/rocp2p_bw_fakeverbs --transport verbs \
    --mr dmabuf --msg 1048576 --region 67108864 --secs 0.3 --json \
    2>&1 | tail -2)
DB2=$(ls gpurun_out/r2/prof_dmabuf/*dmabuf*.db 2>/dev/null | head -1)
[ -n "$DB2" ] && python tools/rocpd_stats.py "$DB2" \
    --out gpurun_out/r2/rocprof_dmabuf_r2.md && \
    tail -12 gpurun_out/r2/rocprof_dmabuf_r2.md

echo "=== native sweep (round-2 tree, both directions) ==="
timeout 300 harness/build/rocp2p_bw --transport hip --sweep --secs 0.4 \
    --json | tee gpurun_out/r2/sweep_write.json
timeout 300 harness/build/rocp2p_bw --transport hip --sweep --secs 0.4 \
    --dir read --json | tee gpurun_out/r2/sweep_read.json

echo "=== 120 s randomized soak (round-2 tree) ==="
timeout 240 python -m rocnrdma_amd.harness.soak --secs 120 \
    2>&1 | tail -4

echo "=== DONE pass2 ==="
