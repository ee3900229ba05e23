#!/bin/bash
# Round-2 GPU call 7: join radix-staged A/B, coarse sweep, SQ PMC on
# the scatter, full gpu suite (incl. min/max lowering).
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call7.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== pytest -m gpu (full) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

echo "=== join A/B: direct vs radix-staged ===" >> $L
timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
BYTEWAX_JOIN_RADIX=1 timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
BYTEWAX_JOIN_RADIX=1 timeout 300 python examples/stream_join_gpu.py >> $L 2>&1

AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"
for cb in 1 2; do
  echo "=== staged coarse=$cb ===" >> $L
  BYTEWAX_SCATTER_COARSE_BITS=$cb timeout 240 python bench.py $AB >> $L 2>&1
done
echo "=== staged rb=12 coarse=1 ===" >> $L
BYTEWAX_SCATTER_COARSE_BITS=1 timeout 240 python bench.py $AB --region-bits 12 >> $L 2>&1

echo "=== SQ PMC on scatter/agg (instr + wait mix) ===" >> $L
export TMPDIR=/tmp
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats --pmc SQ_INSTS_LDS SQ_INSTS_VALU SQ_WAIT_ANY SQ_BUSY_CYCLES \
  -d $REPO/gpurun_out/pmc_sq -o sqpmc -- \
  python $REPO/bench.py --engine native --steps 5 --warmup 2 --batches-per-poll 4 --latency-probes 0 >> $L 2>&1
echo "sq pmc rc=$?" >> $L
cd $REPO
tail -3 $L
