"""Device session windows: per-user activity sessions at GPU scale.

32M events over 1M users; sessions close after a 30s idle gap.
"""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from bytewax_amd.gpu import AGG_COUNT, RecordBatch  # noqa: E402
from bytewax_amd.gpu._ext import ext  # noqa: E402
from bytewax_amd.gpu.state import SessionAggState  # noqa: E402


def main() -> None:
    if not torch.cuda.is_available():
        print("needs a GPU")
        return
    ext()  # warm the extension before timing
    dev = torch.device("cuda:0")
    scratch = torch.empty(8 * (1 << 30), dtype=torch.int8, device=dev)
    del scratch  # pre-touch the allocator (stays cached for reuse)
    n, vocab, gap_ms = 32_000_000, 1_000_000, 30_000
    g = torch.Generator(device="cuda").manual_seed(1)
    st = SessionAggState(dev, gap_ms, AGG_COUNT, slots_pow=21, out_cap=n)
    batches = 8
    keys = [
        torch.randint(0, vocab, (n,), dtype=torch.int32, generator=g,
                      device=dev)
        for _ in range(batches)
    ]
    span = 10_000  # ms of simulated time per batch
    ts0 = torch.randint(0, span, (n,), dtype=torch.int64, generator=g,
                        device=dev)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    closed_sessions = 0
    for i in range(batches):
        # Advance simulated time so per-user gaps occur naturally.
        st.insert(
            RecordBatch(
                keys[i], ts0, None, max_ts=(i + 1) * span - 1,
                ts_base=i * span,
            )
        )
        out = st.close_due()
        if out is not None:
            closed_sessions += len(out["keys"])
    final = st.close_all()
    if final is not None:
        closed_sessions += len(final["keys"])
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    total = n * batches
    print(
        f"sessionized {total} events over {vocab} users in {dt:.2f}s = "
        f"{total / dt:.3g} events/s; {closed_sessions} sessions closed"
    )


if __name__ == "__main__":
    main()
