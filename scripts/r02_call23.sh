#!/bin/bash
# Round-2 call 23: corrected extra datapoints + suite re-confirm.
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_call23.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== 1BRC 100k stations ===" >> $L
sed 's/N_STATIONS = 10_000/N_STATIONS = 100_000/' examples/onebrc_gpu.py > /tmp/onebrc_100k.py
timeout 300 python /tmp/onebrc_100k.py >> $L 2>&1
echo "=== join 1B events (25M x 20 x 2 sides) ===" >> $L
sed 's/events = 10_000_000/events = 25_000_000/' examples/stream_join_gpu.py > /tmp/join_big.py
timeout 300 python /tmp/join_big.py >> $L 2>&1
echo "=== pytest -m gpu (full, re-confirm) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L
tail -4 $L
