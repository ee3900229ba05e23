#!/bin/bash
# Round-2 call 27: A/B pageable vs pinned staging for StringDict H2D
# on the same box (call26 suggested pinned is ~2x slower; confirm).
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_call27.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

timeout 300 python -m pytest tests/test_gpu_strings.py -m gpu -q >> $L 2>&1
echo "canary pytest rc=$?" >> $L
for MODE in 0 1 0 1; do
  echo "=== BYTEWAX_STR_PINNED=$MODE ===" >> $L
  BYTEWAX_STR_PINNED=$MODE timeout 200 python - >> $L 2>&1 <<'PYEOF'
import time, random
import numpy as np, torch
from bytewax_amd.gpu.strings import StringDict, pack_strings

rng = random.Random(7)
vocab = [f"word-{i}" for i in range(200_000)]
N = 4_000_000
packed = pack_strings([vocab[rng.randrange(len(vocab))] for _ in range(N)])
nbytes = packed[0].nbytes + packed[1].nbytes

d = StringDict(torch.device("cuda"), slots_pow=20)
for _ in range(3):
    d.encode(packed)
torch.cuda.synchronize()
t0 = time.perf_counter()
IT = 20
for _ in range(IT):
    d.encode(packed)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"encode: {N*IT/dt/1e9:.2f} G strings/s "
      f"({nbytes*IT/dt/1e9:.1f} GB/s H2D payload)")
PYEOF
done
tail -30 $L
