#!/bin/bash
# Round-2 GPU call 4: validate the fused session path + tiny-cap fix,
# full gpu suite, session/1BRC/join example numbers, staged grid sweep.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call4.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== pytest -m gpu (full) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

echo "=== sessions example (fused path, through run_main) ===" >> $L
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
echo "rc=$?" >> $L
timeout 300 python examples/sessions_gpu.py >> $L 2>&1

echo "=== 1BRC example ===" >> $L
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
echo "=== join example ===" >> $L
timeout 300 python examples/stream_join_gpu.py >> $L 2>&1

AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"
for gb in 256 512 1024 2048; do
  echo "=== staged grid=$gb ===" >> $L
  BYTEWAX_SCATTER_BLOCKS=$gb timeout 240 python bench.py $AB >> $L 2>&1
done

echo "=== headline (dataflow, defaults) ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
echo "rc=$?" >> $L
tail -3 $L
