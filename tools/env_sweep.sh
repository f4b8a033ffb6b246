#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
for sd in 8 16 32; do
  v=$(MA_STAT_DIV=$sd python bench.py --steps 200 --warmup 20 2>/dev/null | tail -1 | python3 -c "import json,sys; print(round(json.load(sys.stdin)['value']))")
  echo "MA_STAT_DIV=$sd -> $v"
done
for cc in 2 4 8; do
  v=$(METRICS_AMD_CURVE_CCHUNK=$cc python bench.py --steps 200 --warmup 20 2>/dev/null | tail -1 | python3 -c "import json,sys; print(round(json.load(sys.stdin)['value']))")
  echo "CURVE_CCHUNK=$cc -> $v"
done
v=$(METRICS_AMD_CURVE_VARIANT=1 python bench.py --steps 200 --warmup 20 2>/dev/null | tail -1 | python3 -c "import json,sys; print(round(json.load(sys.stdin)['value']))")
echo "CURVE_VARIANT=lds -> $v"
for ct in 2 4 8; do
  v=$(MA_SUFFIX_CTILE=$ct python bench.py --steps 200 --warmup 20 2>/dev/null | tail -1 | python3 -c "import json,sys; print(round(json.load(sys.stdin)['value']))")
  echo "SUFFIX_CTILE=$ct -> $v"
done
