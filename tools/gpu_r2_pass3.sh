#!/bin/bash
# Round-2 pass 3: validate the cross-process fabric on real VRAM.
set -u
mkdir -p gpurun_out/r2
exec > >(tee gpurun_out/r2/pass3.log) 2>&1

echo "=== gpu pytest: dmabuf + fabric-into-VRAM tests ==="
timeout 600 python -m pytest tests/test_gpu_native_harness.py -x -q 2>&1 | tail -4

echo "=== 2 writer processes -> one VRAM dmabuf region (aggregate probe) ==="
export FAKE_VERBS_SHM=/rocnr_fab_gpu
harness/build/rocp2p_bw_fakeverbs --serve 0 --mr dmabuf --region 268435456 > gpurun_out/r2/fab_srv.log 2>&1 &
SRV=$!
sleep 3
PORT=$(grep -oP 'listening on port \K\d+' gpurun_out/r2/fab_srv.log)
echo "port=$PORT"
timeout 120 harness/build/rocp2p_bw_fakeverbs --connect 127.0.0.1:$PORT \
    --msg 1048576 --region 268435456 --secs 1.0 | tee gpurun_out/r2/fab_client.json
wait $SRV
rm -f /dev/shm/rocnr_fab_gpu

echo "=== 4 concurrent dmabuf writers, independent regions (aggregate BAR write) ==="
for i in 1 2 3 4; do
  timeout 120 harness/build/rocp2p_bw_fakeverbs --transport verbs \
      --mr dmabuf --msg 1048576 --region 67108864 --secs 1.0 \
      --no-integrity --json > gpurun_out/r2/dmabuf_p$i.json &
done
wait
cat gpurun_out/r2/dmabuf_p*.json

echo "=== DONE pass3 ==="
