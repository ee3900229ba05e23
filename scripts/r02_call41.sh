#!/bin/bash
# Round-2 call 41: extra headline reproducibility datapoints.
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_call41.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
for i in 1 2; do
  echo "=== headline rep $i ===" >> $L
  timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
done
echo "=== examples rep ===" >> $L
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
grep -oE '"value": [0-9.e+]+|sessionized.*|joined.*|aggregated.*' $L | tail -6
