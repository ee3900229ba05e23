"""Property test: the string-exchange CPU twin conserves events and
routes deterministically for arbitrary unicode keys (including empty
strings and duplicate-heavy batches).  Runs the full pack +
collective path as a world-1 gloo self-exchange (`force=True`)."""

import os

import pytest

torch = pytest.importorskip("torch")
hyp = pytest.importorskip("hypothesis")

from hypothesis import given, settings, strategies as st  # noqa: E402

from bytewax_amd.gpu.strings import (  # noqa: E402
    exchange_str_by_key,
    pack_strings,
    str_owner_cpu,
)


@pytest.fixture(scope="module")
def gloo_group():
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29561")
    dist.init_process_group("gloo", rank=0, world_size=1)
    yield dist
    dist.destroy_process_group()


@settings(max_examples=40, deadline=None)
@given(
    words=st.lists(
        st.text(max_size=12), min_size=0, max_size=60
    ),
    with_vals=st.booleans(),
)
def test_self_exchange_conserves_events(gloo_group, words, with_vals):
    data, offs = pack_strings(words)
    n = len(words)
    ts = torch.arange(n, dtype=torch.int64)
    vals = ts * 3 if with_vals else None
    rb, ro, rt, rv = exchange_str_by_key(
        data, offs, ts, vals, force=True
    )
    got = sorted(
        (
            bytes(rb.numpy()[ro[i] : ro[i + 1]]).decode(),
            int(rt[i]),
            int(rv[i]) if rv is not None else None,
        )
        for i in range(len(ro) - 1)
    )
    sent = sorted(
        (w, i, i * 3 if with_vals else None)
        for i, w in enumerate(words)
    )
    assert got == sent


@settings(max_examples=60, deadline=None)
@given(s=st.text(max_size=24), world=st.integers(min_value=1, max_value=8))
def test_owner_is_deterministic_and_in_range(s, world):
    b = s.encode()
    o = str_owner_cpu(b, world)
    assert 0 <= o < world
    assert o == str_owner_cpu(b, world)
