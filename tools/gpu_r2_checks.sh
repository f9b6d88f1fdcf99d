#!/bin/bash
# Round-2 first GPU validation pass.  Run via gpurun from the repo root;
# writes evidence under gpurun_out/.
set -u
mkdir -p gpurun_out/r2
exec > >(tee gpurun_out/r2/first_pass.log) 2>&1

echo "=== box environment probe (kbuild/OFED feasibility) ==="
uname -r
ls /lib/modules/ 2>&1 | head -3
ls /usr/src/ 2>&1 | head -5
ls /usr/include/infiniband 2>&1 | head -2
command -v dpkg >/dev/null && dpkg -l 2>/dev/null | grep -iE "linux-headers|ofed|rdma-core|libibverbs" | head -5
echo "=== amdgpu dmabuf support probe ==="
ls /sys/module/amdgpu/parameters 2>/dev/null | head -3

echo "=== pytest -m gpu ==="
timeout 600 python -m pytest tests -m gpu -x -q 2>&1 | tail -8

echo "=== bench.py sanity (auto -> sdma on this box) ==="
timeout 300 python bench.py --gpus 1 --steps 10 --warmup 3 \
    --json-out gpurun_out/r2/bench_n1.json | tail -2

echo "=== dmabuf MR end-to-end on real VRAM (fakeverbs mmaps the fd) ==="
timeout 120 harness/build/rocp2p_bw_fakeverbs --transport verbs \
    --mr dmabuf --msg 1048576 --region 67108864 --secs 0.3 --json \
    | tee gpurun_out/r2/dmabuf_write.json
timeout 120 harness/build/rocp2p_bw_fakeverbs --transport verbs \
    --mr dmabuf --msg 1048576 --region 67108864 --secs 0.3 --dir read \
    --json | tee gpurun_out/r2/dmabuf_read.json
timeout 180 harness/build/rocp2p_bw_fakeverbs --transport verbs \
    --mr dmabuf --msg 67108864 --region 1073741824 --secs 0.5 --json \
    | tee gpurun_out/r2/dmabuf_write_64m.json

echo "=== N-rank sdma contention on ONE GPU (VERDICT r1 #5) ==="
for n in 2 4; do
  timeout 300 python -m torch.distributed.run --nnodes=1 \
      --nproc-per-node $n --master-addr 127.0.0.1 --master-port 2974$n \
      bench.py --gpus $n --steps 8 --warmup 2 \
      --json-out gpurun_out/r2/contention_${n}rank.json 2>&1 | tail -2
done

echo "=== DONE ==="
