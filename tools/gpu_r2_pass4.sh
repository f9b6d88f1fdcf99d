#!/bin/bash
# Round-2 pass 4: scale + endurance on the round-2 tree.
set -u
mkdir -p gpurun_out/r2
exec > >(tee gpurun_out/r2/pass4.log) 2>&1

echo "=== 16 GiB VRAM dmabuf region (288GB-class sizing on the dmabuf path) ==="
timeout 300 harness/build/rocp2p_bw_fakeverbs --transport verbs \
    --mr dmabuf --msg 67108864 --region 17179869184 --secs 1.0 --json \
    | tee gpurun_out/r2/dmabuf_16g.json

echo "=== bench.py verbs path, 2 ranks, dmabuf MRs on real VRAM ==="
ROCNR_FORCE_VERBS=1 \
ROCNR_VERBS_HARNESS=$PWD/harness/build/rocp2p_bw_fakeverbs \
timeout 300 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29751 \
    bench.py --gpus 2 --steps 5 --warmup 2 --msg-bytes 1048576 \
    --region-bytes 67108864 --verbs-mr dmabuf \
    --json-out gpurun_out/r2/bench_verbs_2rank.json 2>&1 | tail -2

echo "=== 5-minute sustained headline bench ==="
timeout 420 python bench.py --gpus 1 --steps 16000 --warmup 10 \
    --json-out gpurun_out/r2/bench_sustained_r2.json 2>/dev/null | tail -1

echo "=== DONE pass4 ==="
