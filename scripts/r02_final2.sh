#!/bin/bash
# Round-2 final consolidated validation #2 (post join/str/mean work).
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_final2.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== pytest -m gpu (full) ===" >> $L
timeout 1000 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L
echo "=== smoke ===" >> $L
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $L 2>&1
echo "smoke rc=$?" >> $L
echo "=== headline (defaults) ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
echo "=== sessions example (per_poll=4, 1B events) ===" >> $L
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
echo "=== 1BRC example ===" >> $L
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
echo "=== join example ===" >> $L
timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
grep -E "passed|rc=|sessionized|rows/s|joined|\"value\"" $L | tail -12
