"""Deterministic-replay race checks.

The reference delegates race safety to Rust's type system; our
equivalent for the HIP kernels is (a) CPU-twin equality (every gpu
numerics test) and (b) THESE replay checks: integer-atomic
aggregation is order-insensitive, so two runs over the same batches
must produce BITWISE-identical results — any divergence exposes a
data race (e.g. a non-atomic read of a concurrently-written cell),
not just an accuracy bug.
"""

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

from bytewax_amd.gpu import (  # noqa: E402
    AGG_SUM,
    RecordBatch,
    WindowAggState,
)

ALIGN_MS = 1_700_000_000_000


def _mk(seed, n=2_000_000, vocab=50_000):
    g = torch.Generator(device="cuda").manual_seed(seed)
    keys = torch.randint(
        0, vocab, (n,), dtype=torch.int32, device="cuda", generator=g
    )
    ts = torch.randint(
        0, 120_000, (n,), dtype=torch.int64, device="cuda", generator=g
    ) + ALIGN_MS
    vals = torch.randint(
        -50, 50, (n,), dtype=torch.int64, device="cuda", generator=g
    )
    return RecordBatch(keys, ts, vals, max_ts=ALIGN_MS + 119_999)


def _run_once(radix):
    st = WindowAggState(
        torch.device("cuda:0"),
        ALIGN_MS,
        60_000,
        AGG_SUM,
        slots_pow=20,
        out_cap=1 << 20,
        radix=radix,
    )
    for seed in (1, 2, 3):
        st.insert(_mk(seed))
    rows = st.close_all()
    win_idx = (rows.ts - ALIGN_MS) // 60_000  # 0..1 for this span
    order = torch.argsort(rows.keys.to(torch.int64) * 8 + win_idx)
    return (
        rows.keys[order].cpu(),
        rows.ts[order].cpu(),
        rows.vals[order].cpu(),
    )


@pytest.mark.parametrize("radix", [False, True])
def test_window_sum_replay_bitwise_identical(radix):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    a = _run_once(radix)
    b = _run_once(radix)
    for x, y in zip(a, b):
        assert torch.equal(x, y)


def test_dict_encode_replay_consistent():
    """Replaying the same string batch into two dictionaries assigns
    a permutation of ids with identical groupings (id assignment
    order is a benign race; the GROUPING must be exact)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import random

    from bytewax_amd.gpu.strings import StringDict, pack_strings

    rng = random.Random(5)
    words = [f"w-{rng.randrange(30_000)}" for _ in range(1_000_000)]
    packed = pack_strings(words)

    def groups():
        d = StringDict(torch.device("cuda:0"), slots_pow=17)
        ids = d.encode(packed).cpu().tolist()
        return [d.id2str[i] for i in ids]

    assert groups() == groups() == words
