#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
for args in "--batch 2048 --classes 1000" "--batch 32768 --classes 1000" "--batch 8192 --classes 100" "--batch 8192 --classes 5000" "--batch 8192 --classes 1000 --curve-thresholds 1000"; do
  python bench.py --steps 200 --warmup 20 $args 2>/dev/null | tail -1 | python3 tools/_fmt_bench_line.py
done
