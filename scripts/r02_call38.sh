#!/bin/bash
# Round-2 call 38: cProfile the dataflow-engine bench to attribute the
# ~130us/batch Python overhead vs the native loop.
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_call38.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

timeout 420 python - >> $L 2>&1 <<'PYEOF'
import cProfile, pstats, io, sys
sys.argv = ["bench.py", "--steps", "6", "--warmup", "2"]
import bench
pr = cProfile.Profile()
pr.enable()
bench.main()
pr.disable()
sio = io.StringIO()
pstats.Stats(pr, stream=sio).sort_stats("tottime").print_stats(22)
print("\n".join(sio.getvalue().splitlines()[4:34]))
PYEOF
tail -40 $L
