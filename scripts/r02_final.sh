#!/bin/bash
# Round-2 final consolidated validation on one box.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_final.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== pytest -m gpu (full) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L
echo "=== smoke ===" >> $L
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $L 2>&1
echo "=== headline (defaults) ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
echo "=== extended dataflow soak (50 x 200 batches = 640e9 events) ===" >> $L
timeout 600 python bench.py --steps 50 --warmup 5 >> $L 2>&1
echo "=== snapshot/restore rate ===" >> $L
timeout 300 python scripts/measure_snapshot_rate.py >> $L 2>&1
echo "=== examples ===" >> $L
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
tail -5 $L
