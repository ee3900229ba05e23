#!/bin/bash
# Round-2 GPU call 3: full gpu suite (no -x), kernel-stats profile and
# write-amplification PMC with absolute paths.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call3.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== pytest -m gpu (full) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

export TMPDIR=/tmp
cd /tmp
echo "=== rocprofv3 kernel stats (native, staged-c2, 20x10 batches) ===" >> $L
timeout 600 rocprofv3 --kernel-trace --stats -d $REPO/gpurun_out/prof_r02 -o r02stats -- \
  python $REPO/bench.py --engine native --steps 20 --warmup 3 --batches-per-poll 10 --latency-probes 0 >> $L 2>&1
echo "rocprof rc=$?" >> $L
find $REPO/gpurun_out/prof_r02 -type f >> $L 2>&1

echo "=== PMC write amplification (staged) ===" >> $L
timeout 600 rocprofv3 --kernel-trace --stats --pmc TCC_EA0_WRREQ TCC_EA0_WRREQ_64B TCC_EA0_RDREQ \
  -d $REPO/gpurun_out/pmc_r02 -o r02pmc -- \
  python $REPO/bench.py --engine native --steps 5 --warmup 2 --batches-per-poll 4 --latency-probes 0 >> $L 2>&1
echo "pmc rc=$?" >> $L
find $REPO/gpurun_out/pmc_r02 -type f >> $L 2>&1

echo "=== PMC write amplification (fixed, for contrast) ===" >> $L
BYTEWAX_SCATTER=fixed BYTEWAX_SCATTER_COARSE_BITS=0 timeout 600 rocprofv3 --kernel-trace --stats \
  --pmc TCC_EA0_WRREQ TCC_EA0_WRREQ_64B TCC_EA0_RDREQ \
  -d $REPO/gpurun_out/pmc_r02f -o r02pmcf -- \
  python $REPO/bench.py --engine native --steps 5 --warmup 2 --batches-per-poll 4 --latency-probes 0 >> $L 2>&1
echo "pmcf rc=$?" >> $L

tail -3 $L
