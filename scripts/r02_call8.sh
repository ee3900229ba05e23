#!/bin/bash
# Round-2 GPU call 8: measure the LDS-latency batching changes.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call8.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== gpu numerics (twin + kernels + lowering) ===" >> $L
timeout 900 python -m pytest tests/test_gpu_cpu_twin.py tests/test_gpu_kernels.py tests/windowing/test_columnar_lowering.py tests/test_gpu_stats_join.py tests/test_gpu_sessions.py -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"
echo "=== native after batching ===" >> $L
timeout 240 python bench.py $AB >> $L 2>&1
timeout 240 python bench.py $AB >> $L 2>&1
echo "=== headline (dataflow, B=20) ===" >> $L
timeout 240 python bench.py --steps 10 --warmup 3 --batches-per-poll 20 >> $L 2>&1
echo "=== kernel stats ===" >> $L
export TMPDIR=/tmp; cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d $REPO/gpurun_out/prof_r02b -o r02b -- \
  python $REPO/bench.py --engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0 >> $L 2>&1
cd $REPO
tail -3 $L
