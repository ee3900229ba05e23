#!/bin/bash
# Round-2 call 37: full suite (incl. new determinism tests) + mega
# soak (6.4e12 events native; 100-step dataflow soak) + 1BRC re-measure.
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_call37.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== pytest -m gpu (full, incl. determinism) ===" >> $L
timeout 1000 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

echo "=== native mega soak: 500 steps x 200 batches x 64M = 6.4e12 events ===" >> $L
timeout 600 python bench.py --engine native --steps 500 --warmup 5 \
  --batches-per-poll 200 --latency-probes 20 >> $L 2>&1
echo "rc=$?" >> $L

echo "=== dataflow soak: 100 steps (1.28e12 events) ===" >> $L
timeout 600 python bench.py --steps 100 --warmup 5 >> $L 2>&1
echo "rc=$?" >> $L

echo "=== 1BRC (per_poll=4) x2 ===" >> $L
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
grep -E "passed|failed|rc=|rows/s" $L | tail -8
grep -o '"value": [0-9.e+]*\|"p99_step_ms": [0-9.]*\|closed_window_rows": [0-9]*' $L >> $L
tail -14 $L
