#!/bin/bash
# Round-2 GPU call 1: validate the staged scatter + new dataflow-engine
# headline, A/B scatter variants, probe 2-rank RCCL on one GPU.
set -x
cd /root/repo
mkdir -p gpurun_out
L=gpurun_out/r02_call1.log
: > $L

echo "=== pytest -m gpu ===" >> $L
timeout 900 python -m pytest tests -m gpu -x -q >> $L 2>&1
echo "pytest rc=$?" >> $L

echo "=== headline bench: dataflow engine, defaults ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
echo "rc=$?" >> $L

AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"
for v in staged fixed direct; do
  echo "=== native A/B scatter=$v ===" >> $L
  BYTEWAX_SCATTER=$v timeout 240 python bench.py $AB >> $L 2>&1
done
echo "=== native A/B staged coarse=0 ===" >> $L
BYTEWAX_SCATTER=staged BYTEWAX_SCATTER_COARSE_BITS=0 timeout 240 python bench.py $AB >> $L 2>&1
echo "=== native A/B staged coarse=3 ===" >> $L
BYTEWAX_SCATTER=staged BYTEWAX_SCATTER_COARSE_BITS=3 timeout 240 python bench.py $AB >> $L 2>&1
echo "=== native A/B fixed coarse=2 ===" >> $L
BYTEWAX_SCATTER=fixed BYTEWAX_SCATTER_COARSE_BITS=2 timeout 240 python bench.py $AB >> $L 2>&1

echo "=== dataflow engine A/B staged vs fixed (short) ===" >> $L
BYTEWAX_SCATTER=staged timeout 240 python bench.py --steps 10 --warmup 3 --batches-per-poll 10 >> $L 2>&1
BYTEWAX_SCATTER=fixed timeout 240 python bench.py --steps 10 --warmup 3 --batches-per-poll 10 >> $L 2>&1

echo "=== 2-rank RCCL probe (one GPU) ===" >> $L
timeout 240 python scripts/probe_nccl_2rank.py 2 >> $L 2>&1
echo "probe rc=$?" >> $L

tail -5 $L
