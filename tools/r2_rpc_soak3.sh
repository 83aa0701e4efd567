#!/bin/bash
# Final soak config: 3 workers x 3 contexts, GC-tuned server + GC-disabled
# clients, 4 clients x 200 req/s = 800 req/s total.
cd /root/repo
mkdir -p gpurun_out
timeout 400 python examples/inference_server.py --workers 3 --contexts 3 \
  --port 50954 --metrics-port 51278 > gpurun_out/rpc3_server.log 2>&1 &
SRV=$!
sleep 45
echo "=== 60 s soak @ 800 req/s total (4 clients x 200), pure protobuf ==="
PIDS=""
for i in a b c d; do
  timeout 150 python examples/siege.py --target 127.0.0.1:50954 --rate 200 \
    --seconds 60 > gpurun_out/rpc3_soak_$i.log 2>&1 &
  PIDS="$PIDS $!"
done
wait $PIDS
for i in a b c d; do echo "--- client $i ---"; cat gpurun_out/rpc3_soak_$i.log; done
kill $SRV 2>/dev/null; wait $SRV 2>/dev/null
echo "=== done ==="
