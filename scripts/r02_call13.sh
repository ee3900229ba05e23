#!/bin/bash
# Round-2 call 13: U sweep + numerics check.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call13.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== numerics (twin subset) ===" >> $L
timeout 600 python -m pytest tests/test_gpu_cpu_twin.py tests/test_gpu_kernels.py -m gpu -q >> $L 2>&1
echo "rc=$?" >> $L
AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"
for u in 8 16 32; do
  echo "=== U=$u ===" >> $L
  BYTEWAX_SCATTER_U=$u timeout 240 python bench.py $AB >> $L 2>&1
  BYTEWAX_SCATTER_U=$u timeout 240 python bench.py $AB >> $L 2>&1
done
tail -3 $L
