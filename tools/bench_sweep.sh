#!/bin/bash
# Full model x dtype sweep; prints one summary line per config.
cd "$(dirname "$0")/.."
for cfg in "resnet50 fp16" "resnet50 int8" "resnet50 fp8" \
           "resnet152 fp16" "resnet152 int8" "resnet152 fp8" \
           "bert fp16" "bert fp8" "vit fp16" "llama fp16" "gpt2 fp16"; do
  set -- $cfg
  timeout 250 python bench.py --model "$1" --dtype "$2" --steps 300 --warmup 30 \
    2>"gpurun_out/sweep_err_$1_$2.log" | python tools/fmt_bench.py
done
