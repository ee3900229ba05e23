#!/bin/bash
# Round-2 GPU call 5: stats staged-scatter validation + 1BRC A/B.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call5.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== pytest -m gpu (full) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

echo "=== 1BRC staged (default) x2 ===" >> $L
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
echo "=== 1BRC fixed (r1 path) ===" >> $L
BYTEWAX_SCATTER=fixed BYTEWAX_SCATTER_COARSE_BITS=0 timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
echo "=== 1BRC 100k stations (staged) ===" >> $L
timeout 300 python examples/onebrc_gpu.py --rows 1000000000 >> $L 2>&1

echo "=== sessions rep ===" >> $L
timeout 300 python examples/sessions_gpu.py >> $L 2>&1

echo "=== headline confirm (short) ===" >> $L
timeout 240 python bench.py --steps 10 --warmup 3 --batches-per-poll 20 >> $L 2>&1
tail -3 $L
