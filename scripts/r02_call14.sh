#!/bin/bash
# Round-2 call 14: staged2 A/B + correctness.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call14.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== staged2 numerics (twin subset, forced via env) ===" >> $L
BYTEWAX_SCATTER=staged2 timeout 600 python -m pytest tests/test_gpu_cpu_twin.py tests/test_gpu_kernels.py -m gpu -q >> $L 2>&1
echo "rc=$?" >> $L
AB="--engine native --no-pipeline --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0"
for v in staged staged2; do
  echo "=== no-pipeline $v ===" >> $L
  BYTEWAX_SCATTER=$v timeout 240 python bench.py $AB >> $L 2>&1
  BYTEWAX_SCATTER=$v timeout 240 python bench.py $AB >> $L 2>&1
done
echo "=== kernel stats staged2 ===" >> $L
export TMPDIR=/tmp; cd /tmp
BYTEWAX_SCATTER=staged2 timeout 600 rocprofv3 --kernel-trace --stats -d $REPO/gpurun_out/prof_s2 -o s2 -- \
  python $REPO/bench.py --engine native --no-pipeline --steps 10 --warmup 3 --batches-per-poll 10 --latency-probes 0 >> $L 2>&1
cd $REPO
tail -3 $L
