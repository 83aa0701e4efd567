#!/bin/bash
# Round-2 check 1: comm layer proof + dp2-on-1-GPU bench + replica serving.
cd /root/repo
mkdir -p gpurun_out
echo "=== gpu probe ==="
timeout 120 python tools/gpu_probe.py 2>&1 | tail -3

echo "=== comm gpu tests ==="
timeout 900 python -m pytest tests/test_comm_gpu.py -q -m gpu 2>&1 | tail -5 | tee gpurun_out/comm_test.log

echo "=== dp2 bench on one GPU (2 RCCL ranks share device 0) ==="
timeout 600 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29741 bench.py --gpus 2 --steps 100 \
  --warmup 10 2>&1 | tail -6 | tee gpurun_out/bench_dp2_1gpu.log

echo "=== 1-GPU bench baseline (post-refactor regression check) ==="
timeout 600 python bench.py --steps 300 --warmup 30 2>&1 | tail -2 | tee gpurun_out/bench_1gpu_r2a.log

echo "=== replica serving: two engines on device 0 under siege load ==="
timeout 300 python examples/inference_server.py --devices 0,0 --contexts 2 \
  --metrics-port 50978 --port 50951 > gpurun_out/replica_server.log 2>&1 &
SRV=$!
sleep 25
timeout 120 python examples/siege.py --target 127.0.0.1:50951 --rate 400 \
  --seconds 10 2>&1 | tee gpurun_out/replica_siege.log
kill $SRV 2>/dev/null
wait $SRV 2>/dev/null
tail -5 gpurun_out/replica_server.log
echo "=== done ==="
