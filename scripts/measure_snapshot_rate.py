"""Measure device-state recovery spill/restore rates (checkpoint
subsystem evidence): extract-compact + pinned-host copy of a
many-cell HBM window table, and the rebuild on restore."""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from bytewax_amd.gpu import AGG_COUNT, RecordBatch, WindowAggState


def main():
    dev = torch.device("cuda:0")
    n_cells = 16_000_000
    st = WindowAggState(
        dev, 0, 60_000, AGG_COUNT, slots_pow=25, out_cap=n_cells + 1024,
        radix=True, region_bits=12, max_batch=n_cells,
    )
    g = torch.Generator(device="cuda").manual_seed(9)
    keys = torch.randint(
        0, n_cells, (n_cells,), dtype=torch.int32, generator=g, device=dev
    )
    ts = torch.randint(
        0, 600_000, (n_cells,), dtype=torch.int64, generator=g, device=dev
    )
    st.insert(RecordBatch(keys, ts, max_ts=600_000))
    torch.cuda.synchronize()

    t0 = time.perf_counter()
    snap = st.snapshot_to_host()
    t_spill = time.perf_counter() - t0
    rows = len(snap["keys"])
    nbytes = rows * 16  # key4 + win4 + val8

    st2 = WindowAggState(
        dev, 0, 60_000, AGG_COUNT, slots_pow=25, out_cap=n_cells + 1024,
        radix=True, region_bits=12, max_batch=n_cells,
    )
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    st2.restore_from_host(snap)
    torch.cuda.synchronize()
    t_rest = time.perf_counter() - t0

    check = st2.close_all()
    print(
        f"snapshot: {rows} live cells ({nbytes / 1e6:.0f} MB) spilled "
        f"to pinned host in {t_spill * 1000:.1f} ms = "
        f"{rows / t_spill / 1e6:.0f}M cells/s ({nbytes / t_spill / 1e9:.1f} GB/s); "
        f"restore {t_rest * 1000:.1f} ms = {rows / t_rest / 1e6:.0f}M cells/s; "
        f"restored rows verify: {len(check)} cells, "
        f"sum {int(check.vals.sum())}"
    )


if __name__ == "__main__":
    main()
