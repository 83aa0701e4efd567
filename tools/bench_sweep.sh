#!/bin/bash
# Full model x dtype sweep; prints one summary line per config.
for cfg in "resnet50 fp16" "resnet50 int8" "resnet50 fp8" \
           "resnet152 fp16" "resnet152 int8" "resnet152 fp8" \
           "bert fp16" "bert fp8"; do
  set -- $cfg
  out=$(timeout 250 python bench.py --model $1 --dtype $2 --steps 300 --warmup 30 2>/dev/null | tail -1)
  echo "$out" | python - <<'PYEOF'
import json, sys
line = sys.stdin.read().strip()
try:
    d = json.loads(line)
    c = d["config"]
    print(f'{c["model"]:>10} {d["dtype"]:>5}: {d["value"]:>9} inf/s  '
          f'{d["ms_per_step"]:>7} ms/step  p99={c["p99_ms"]:.2f} ms')
except Exception as e:
    print("PARSE FAIL:", line[:100], e)
PYEOF
done
