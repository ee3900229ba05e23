#!/bin/bash
# Round-2 final confirm on the as-shipped tree.
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_final3.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== pytest -m gpu (full) ===" >> $L
timeout 1000 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L
echo "=== smoke ===" >> $L
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $L 2>&1
echo "smoke rc=$?" >> $L
echo "=== headline (driver defaults) ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
echo "rc=$?" >> $L
grep -E "passed|rc=|\"value\"" $L | tail -6
