#!/bin/bash
# Round-2 call 36: new NCCL str-exchange test + amortized join example
# + suite re-confirm.
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_call36.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== new/changed gpu tests ===" >> $L
timeout 600 python -m pytest tests/test_exchange_nccl.py tests/test_gpu_strings.py tests/test_gpu_stats_join.py -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

echo "=== join example (per_poll=4, 800M events) x2 ===" >> $L
timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
echo "=== sessions re-confirm ===" >> $L
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
grep -E "passed|failed|rc=|joined|sessionized" $L | tail -8
