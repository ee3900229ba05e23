"""MI355X columnar tumbling-window wordcount.

The GPU-native version of the flagship workload: synthetic keyed
events in HBM, fused HIP window-count insert, RCCL all-to-allv key
exchange when run with multiple ranks.

1 GPU:  python examples/gpu_wordcount.py
N GPUs: python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
            --master-addr 127.0.0.1 examples/gpu_wordcount.py
"""

import os
import sys
from datetime import datetime, timedelta, timezone
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.gpu.operators import (
    CollectCountsSink,
    SyntheticEventSource,
    keyed_window_agg,
)
from bytewax_amd.testing import run_main


def main():
    import torch

    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        import torch.distributed as dist

        dist.init_process_group(
            backend="nccl" if torch.cuda.is_available() else "gloo"
        )
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    device = "cuda" if torch.cuda.is_available() else "cpu"

    align = datetime(2024, 1, 1, tzinfo=timezone.utc)
    out = []
    flow = Dataflow("gpu_wordcount")
    s = op.input(
        "inp",
        flow,
        SyntheticEventSource(
            events_per_batch=1_000_000,
            n_batches=20,
            vocab=100_000,
            align_to=align,
            sim_ms_per_batch=10_000,
            device=device,
        ),
    )
    agg = keyed_window_agg(
        "window_count",
        s,
        align_to=align,
        length=timedelta(minutes=1),
        mode="count",
        device=device,
    )
    op.output("out", agg, CollectCountsSink(out))
    run_main(flow)
    total = sum(int(b.vals.sum().item()) for b in out)
    print(f"counted {total} events into {sum(len(b) for b in out)} "
          "(word, window) cells")


if __name__ == "__main__":
    main()
