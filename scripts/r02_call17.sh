#!/bin/bash
# Round-2 call 17: grid-1024 defaults — numerics + dataflow pipe A/B.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call17.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== numerics (twins + kernels + sessions + stats) ===" >> $L
timeout 900 python -m pytest tests/test_gpu_cpu_twin.py tests/test_gpu_kernels.py tests/test_gpu_sessions.py tests/test_gpu_stats_join.py tests/windowing/test_columnar_lowering.py -m gpu -q >> $L 2>&1
echo "rc=$?" >> $L
AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"
echo "=== native serial (grid-1024 default) x2 ===" >> $L
timeout 240 python bench.py $AB >> $L 2>&1
timeout 240 python bench.py $AB >> $L 2>&1
DF="--steps 20 --warmup 5"
echo "=== dataflow default (pipe) ===" >> $L
timeout 420 python bench.py $DF >> $L 2>&1
echo "=== dataflow serial insert ===" >> $L
BYTEWAX_PY_PIPELINE=0 timeout 420 python bench.py $DF >> $L 2>&1
BYTEWAX_PY_PIPELINE=0 timeout 420 python bench.py $DF >> $L 2>&1
echo "=== examples quick ===" >> $L
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
tail -3 $L
