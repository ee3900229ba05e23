#!/bin/bash
# Round-2 call 18: trillion-event soak + full suite on final defaults.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call18.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== pytest -m gpu (full) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L
echo "=== smoke ===" >> $L
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $L 2>&1
echo "=== TRILLION-event native soak (2000 steps x 10 batches x 64M) ===" >> $L
timeout 900 python bench.py --engine native --steps 2000 --warmup 5 --batches-per-poll 10 --latency-probes 20 >> $L 2>&1
echo "=== dataflow soak (50 x 200) ===" >> $L
timeout 600 python bench.py --steps 50 --warmup 5 >> $L 2>&1
echo "=== headline (defaults) ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
tail -3 $L
