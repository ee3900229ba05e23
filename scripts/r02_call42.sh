#!/bin/bash
# Round-2 call 42: host-path config-1 re-measure after the hot-loop
# work + CPU suite sanity on the box.
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_call42.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== host config-1 bench (best of 7) ===" >> $L
timeout 300 python scripts/host_bench.py >> $L 2>&1
echo "=== engine/operator tests on the box ===" >> $L
timeout 600 python -m pytest tests/operators tests/test_dataflow.py tests/test_recovery.py -q >> $L 2>&1
echo "rc=$?" >> $L
tail -5 $L
