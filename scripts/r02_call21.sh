#!/bin/bash
# Round-2 call 21: the definitive final record on the final defaults.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call21.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== pytest -m gpu (full) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L
echo "=== smoke ===" >> $L
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $L 2>&1
echo "=== TRILLION-event native soak ===" >> $L
timeout 900 python bench.py --engine native --steps 2000 --warmup 5 --batches-per-poll 10 --latency-probes 20 >> $L 2>&1
echo "=== dataflow soak 50x200 ===" >> $L
timeout 600 python bench.py --steps 50 --warmup 5 >> $L 2>&1
echo "=== headline x2 ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
echo "=== examples (final numbers) ===" >> $L
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
echo "=== snapshot rate ===" >> $L
timeout 300 python scripts/measure_snapshot_rate.py >> $L 2>&1
echo "=== final kernel stats for the record ===" >> $L
export TMPDIR=/tmp; cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d $REPO/gpurun_out/prof_final2 -o fin2 -- \
  python $REPO/bench.py --engine native --steps 20 --warmup 3 --batches-per-poll 10 --latency-probes 0 >> $L 2>&1
cd $REPO
tail -3 $L
