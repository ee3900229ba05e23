#!/bin/bash
# Round-2 call 19: block-size + coarse-3 occupancy sweep (serial native).
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call19.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== numerics quick ===" >> $L
BYTEWAX_SCATTER_THREADS=512 timeout 600 python -m pytest tests/test_gpu_cpu_twin.py -m gpu -q >> $L 2>&1
echo "rc=$?" >> $L
AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"
echo "=== baseline (256 thr, grid 1024) ===" >> $L
timeout 240 python bench.py $AB >> $L 2>&1
echo "=== 512 thr ===" >> $L
BYTEWAX_SCATTER_THREADS=512 timeout 240 python bench.py $AB >> $L 2>&1
BYTEWAX_SCATTER_THREADS=512 BYTEWAX_SCATTER_BLOCKS=512 timeout 240 python bench.py $AB >> $L 2>&1
echo "=== 1024 thr ===" >> $L
BYTEWAX_SCATTER_THREADS=1024 BYTEWAX_SCATTER_BLOCKS=512 timeout 240 python bench.py $AB >> $L 2>&1
echo "=== rb10 coarse3 (nseg 256) ===" >> $L
BYTEWAX_SCATTER_COARSE_BITS=3 timeout 240 python bench.py $AB --region-bits 10 >> $L 2>&1
BYTEWAX_SCATTER_COARSE_BITS=3 BYTEWAX_SCATTER_BLOCKS=2048 timeout 240 python bench.py $AB --region-bits 10 >> $L 2>&1
tail -3 $L
