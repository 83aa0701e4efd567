#!/bin/bash
# Soak rerun: 800 req/s split across 2 siege processes (the previous run's
# p99 was CLIENT-side queueing: one python process serializing ~1 GB/s).
cd /root/repo
mkdir -p gpurun_out
timeout 400 python examples/inference_server.py --workers 3 --contexts 2 \
  --port 50953 --metrics-port 51178 > gpurun_out/rpc2_server.log 2>&1 &
SRV=$!
sleep 45
echo "=== 60 s soak @ 800 req/s total (2 clients x 400), pure protobuf ==="
timeout 150 python examples/siege.py --target 127.0.0.1:50953 --rate 400 \
  --seconds 60 > gpurun_out/rpc2_soak_a.log 2>&1 &
A=$!
timeout 150 python examples/siege.py --target 127.0.0.1:50953 --rate 400 \
  --seconds 60 > gpurun_out/rpc2_soak_b.log 2>&1 &
B=$!
wait $A $B
echo "--- client A ---"; cat gpurun_out/rpc2_soak_a.log
echo "--- client B ---"; cat gpurun_out/rpc2_soak_b.log
kill $SRV 2>/dev/null; wait $SRV 2>/dev/null
echo "=== done ==="
