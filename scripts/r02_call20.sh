#!/bin/bash
# Round-2 call 20: validate 512-thread defaults end to end.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call20.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== pytest -m gpu (full) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L
echo "=== headline (defaults) x2 ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
echo "=== native serial x2 ===" >> $L
AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 20"
timeout 240 python bench.py $AB >> $L 2>&1
timeout 240 python bench.py $AB >> $L 2>&1
echo "=== examples ===" >> $L
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
tail -3 $L
