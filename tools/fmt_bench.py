import json
import sys

line = sys.stdin.read().strip().splitlines()
line = line[-1] if line else ""
try:
    d = json.loads(line)
    c = d["config"]
    print(f'{c["model"]:>10} {d["dtype"]:>5}: {d["value"]:>9} inf/s  '
          f'{d["ms_per_step"]:>7} ms/step  p99={c["p99_ms"]:.2f} ms')
except Exception as e:
    print("PARSE FAIL:", repr(line)[:160], e)
