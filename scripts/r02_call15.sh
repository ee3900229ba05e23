#!/bin/bash
# Round-2 call 15: pipeline-vs-serial A/B on one box, both engines.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call15.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"
echo "=== native pipelined ===" >> $L
timeout 240 python bench.py $AB >> $L 2>&1
timeout 240 python bench.py $AB >> $L 2>&1
echo "=== native no-pipeline ===" >> $L
timeout 240 python bench.py $AB --no-pipeline >> $L 2>&1
timeout 240 python bench.py $AB --no-pipeline >> $L 2>&1
DF="--steps 10 --warmup 3 --batches-per-poll 20"
echo "=== dataflow pipelined ===" >> $L
timeout 240 python bench.py $DF >> $L 2>&1
echo "=== dataflow no-pipeline (BYTEWAX_PY_PIPELINE=0) ===" >> $L
BYTEWAX_PY_PIPELINE=0 timeout 240 python bench.py $DF >> $L 2>&1
BYTEWAX_PY_PIPELINE=0 timeout 240 python bench.py $DF >> $L 2>&1
echo "=== staged2 post-fix numerics ===" >> $L
BYTEWAX_SCATTER=staged2 timeout 600 python -m pytest tests/test_gpu_cpu_twin.py -m gpu -q >> $L 2>&1
echo "rc=$?" >> $L
tail -3 $L
