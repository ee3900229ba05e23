#!/bin/bash
# Round-2 call 33: SC_GRAN=16 experiment (128 B cursor granules) on
# the headline path.  Patches the BOX COPY of the kernel source and
# rebuilds there — the snapshot is discarded, repo unaffected.
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_call33.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"

echo "=== baseline (SC_GRAN=8, shipped .so) ===" >> $L
timeout 240 python bench.py $AB >> $L 2>&1

echo "=== patch SC_GRAN=16 + rebuild on box ===" >> $L
sed -i 's/#define SC_GRAN 8 /#define SC_GRAN 16 /' bytewax_amd/_native/stream_kernels.hip
grep -n "define SC_GRAN" bytewax_amd/_native/stream_kernels.hip >> $L
timeout 600 python -c "import __graft_entry__; __graft_entry__.build()" >> $L 2>&1
echo "build rc=$?" >> $L

echo "=== SC_GRAN=16 native bench ===" >> $L
timeout 240 python bench.py $AB >> $L 2>&1
echo "=== SC_GRAN=16 + 1024 scatter threads ===" >> $L
BYTEWAX_SCATTER_THREADS=1024 timeout 240 python bench.py $AB >> $L 2>&1
echo "=== SC_GRAN=16 repeat ===" >> $L
timeout 240 python bench.py $AB >> $L 2>&1
grep -o '"value": [0-9.e+]*' $L >> $L
tail -8 $L
