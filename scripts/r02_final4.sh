#!/bin/bash
# Round-2 final confirm #4: after the host-engine hot-loop pass
# (shared stateful_batch executor drives the GPU path too).
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_final4.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== pytest -m gpu (full) ===" >> $L
timeout 1000 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L
echo "=== smoke ===" >> $L
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $L 2>&1
echo "smoke rc=$?" >> $L
echo "=== headline ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
grep -E "passed|rc=|\"value\"" $L | tail -5
