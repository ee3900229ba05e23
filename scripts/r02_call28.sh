#!/bin/bash
# Round-2 call 28: LDS-deduped region join — correctness + 3-way A/B
# (direct atomics vs radix+LDS vs radix without LDS).
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_call28.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== join + stats gpu tests (exercises radix=True -> LDS kernel) ===" >> $L
timeout 400 python -m pytest tests/test_gpu_stats_join.py -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

for CFG in "BYTEWAX_JOIN_RADIX=0 # direct" \
           "BYTEWAX_JOIN_RADIX=1 # radix+LDS" \
           "BYTEWAX_JOIN_RADIX=1 BYTEWAX_JOIN_LDS=0 # radix no-LDS"; do
  ENVS=${CFG%%#*}
  echo "=== $CFG ===" >> $L
  env $ENVS timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
done
echo "=== radix+LDS repeat ===" >> $L
BYTEWAX_JOIN_RADIX=1 timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
tail -30 $L
