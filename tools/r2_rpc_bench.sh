#!/bin/bash
# Multi-worker (SO_REUSEPORT) protobuf service measurement — VERDICT item 6.
cd /root/repo
mkdir -p gpurun_out
timeout 500 python examples/inference_server.py --workers 3 --contexts 2 \
  --port 50952 --metrics-port 51078 > gpurun_out/rpc_server.log 2>&1 &
SRV=$!
sleep 45
echo "=== 4 parallel sync clients (separate processes) x 3 workers ==="
for i in 1 2 3 4; do
  timeout 120 python examples/client.py --target 127.0.0.1:50952 \
    --count 400 --mode sync > gpurun_out/rpc_sync_$i.log 2>&1 &
done
wait %2 %3 %4 %5 2>/dev/null
grep -h "inf/sec" gpurun_out/rpc_sync_*.log
echo "=== async client ==="
timeout 120 python examples/client.py --target 127.0.0.1:50952 --count 1200 \
  --mode async 2>&1 | tail -2 | tee gpurun_out/rpc_async.log
echo "=== 60 s soak @ 800 req/s, pure protobuf (no shm) ==="
timeout 150 python examples/siege.py --target 127.0.0.1:50952 --rate 800 \
  --seconds 60 2>&1 | tee gpurun_out/rpc_soak800.log
kill $SRV 2>/dev/null
wait $SRV 2>/dev/null
tail -3 gpurun_out/rpc_server.log
echo "=== done ==="
