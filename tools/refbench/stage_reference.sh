#!/bin/bash
# Stage the reference torchmetrics source (read-only mount) for baseline
# benchmarking. The copy lives only in this gitignored directory and is
# removed after measurement — it is never committed.
mkdir -p "$(dirname "$0")/_staged"
cp -r /root/reference/src/torchmetrics "$(dirname "$0")/_staged/torchmetrics"
