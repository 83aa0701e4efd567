#!/usr/bin/env python3
"""Convert a rocprofv3 kernel-trace CSV into chrome://tracing JSON
(aux tracing subsystem: per-dispatch GPU timeline you can scrub in
chrome://tracing or Perfetto, lanes = queues).

    rocprofv3 --kernel-trace --output-format csv -d out -o run -- <cmd>
    python tools/trace2chrome.py out/run_kernel_trace.csv trace.json
"""
import csv
import json
import sys


def _col(row, *names, default=None):
    for n in names:
        if n in row and row[n] != "":
            return row[n]
    return default


def convert(csv_path: str, out_path: str, max_name: int = 120) -> int:
    """Returns the number of events written. Tolerant of rocprofv3
    column-name variants across versions."""
    events = []
    with open(csv_path) as f:
        for row in csv.DictReader(f):
            name = _col(row, "Kernel_Name", "Name", "KernelName")
            t0 = _col(row, "Start_Timestamp", "BeginNs", "Start_Time")
            t1 = _col(row, "End_Timestamp", "EndNs", "End_Time")
            if not (name and t0 and t1):
                continue
            q = _col(row, "Queue_Id", "Queue-Id", "queue-id", default="0")
            a = _col(row, "Agent_Id", "GPU-Id", "gpu-id", default="0")
            t0, t1 = int(t0), int(t1)
            events.append({
                "name": name[:max_name],
                "ph": "X",                    # complete event
                "ts": t0 / 1e3,               # ns -> us
                "dur": max(t1 - t0, 1) / 1e3,
                "pid": f"gpu{a}",
                "tid": f"queue{q}",
                "cat": "kernel",
            })
    with open(out_path, "w") as f:
        json.dump({"traceEvents": events,
                   "displayTimeUnit": "ms"}, f)
    return len(events)


if __name__ == "__main__":
    if len(sys.argv) != 3:
        raise SystemExit(__doc__)
    n = convert(sys.argv[1], sys.argv[2])
    print(f"{n} events -> {sys.argv[2]}")
