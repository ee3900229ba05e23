#!/bin/bash
# Round-2 GPU call 2: int32-ts validation, headline rerun, scatter A/B
# on the final kernels, kernel-stats profile + write-amplification PMC.
set -x
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
L=gpurun_out/r02_call2.log
: > $L

echo "=== pytest -m gpu ===" >> $L
timeout 1200 python -m pytest tests -m gpu -x -q >> $L 2>&1
echo "pytest rc=$?" >> $L

echo "=== headline bench: dataflow engine, defaults (B=200, int32 ts) ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
echo "rc=$?" >> $L

AB="--engine native --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"
for v in staged fixed; do
  for cb in 2 3; do
    echo "=== native A/B scatter=$v coarse=$cb (int32 ts) ===" >> $L
    BYTEWAX_SCATTER=$v BYTEWAX_SCATTER_COARSE_BITS=$cb timeout 240 python bench.py $AB >> $L 2>&1
  done
done

echo "=== rocprofv3 kernel stats (native, staged, 20x10 batches) ===" >> $L
cd /tmp && rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_r02 -o r02stats -- \
  python /root/repo/bench.py --engine native --steps 20 --warmup 3 --batches-per-poll 10 --latency-probes 0 >> $L 2>&1
echo "rocprof rc=$?" >> $L
cd /root/repo
find gpurun_out/prof_r02 -name "*stats*" | head >> $L

echo "=== PMC write amplification (staged scatter) ===" >> $L
cd /tmp && rocprofv3 --kernel-trace --stats --pmc TCC_EA0_WRREQ TCC_EA0_WRREQ_64B TCC_EA0_RDREQ \
  -d /root/repo/gpurun_out/pmc_r02 -o r02pmc -- \
  python /root/repo/bench.py --engine native --steps 5 --warmup 2 --batches-per-poll 4 --latency-probes 0 >> $L 2>&1
echo "pmc rc=$?" >> $L
cd /root/repo
ls gpurun_out/pmc_r02 >> $L 2>&1

tail -5 $L
