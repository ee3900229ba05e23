#!/bin/bash
# Round-2 call 16: validation with serial-native default + serial
# grid/U sweeps + headline re-confirm.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call16.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== pytest -m gpu (full) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L
echo "=== headline (dataflow, defaults) ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"
echo "=== native serial (new default) ===" >> $L
timeout 240 python bench.py $AB >> $L 2>&1
for gb in 1024 2048; do
  echo "=== serial grid=$gb ===" >> $L
  BYTEWAX_SCATTER_BLOCKS=$gb timeout 240 python bench.py $AB >> $L 2>&1
done
echo "=== serial U=8 ===" >> $L
BYTEWAX_SCATTER_U=8 timeout 240 python bench.py $AB >> $L 2>&1
echo "=== dataflow serial-insert (BYTEWAX_PY_PIPELINE=0) rep ===" >> $L
BYTEWAX_PY_PIPELINE=0 timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
echo "=== examples ===" >> $L
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
tail -3 $L
