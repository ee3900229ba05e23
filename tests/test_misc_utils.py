"""Small utility parity tests (reference inline Rust tests analog)."""

from datetime import timedelta

from bytewax_amd._utils import partition
from bytewax_amd.operators.helpers import map_dict_value
from bytewax_amd.recovery import _epochs_per, _route_part


def test_partition():
    trues, falses = partition(range(6), lambda x: x % 2 == 0)
    assert trues == [0, 2, 4]
    assert falses == [1, 3, 5]


def test_map_dict_value():
    f = map_dict_value("name", str.upper)
    assert f({"name": "ada", "x": 1}) == {"name": "ADA", "x": 1}


def test_epochs_per():
    """Reference inputs.rs:79-91 semantics: epochs fully covering a
    duration."""
    i = timedelta(seconds=10)
    assert _epochs_per(timedelta(seconds=0), i) == 0
    assert _epochs_per(timedelta(seconds=10), i) == 1
    assert _epochs_per(timedelta(seconds=15), i) == 2
    assert _epochs_per(timedelta(seconds=20), i) == 2
    assert _epochs_per(timedelta(seconds=25), i) == 3


def test_route_part_stable():
    # Partition routing must be stable across processes/runs.
    assert _route_part("step", "key", 7) == _route_part("step", "key", 7)
    spread = {_route_part("s", f"k{i}", 8) for i in range(100)}
    assert len(spread) == 8


def test_ttl_cache_expires_entries():
    from datetime import datetime, timedelta, timezone

    from bytewax_amd.operators import TTLCache

    t = [datetime(2024, 1, 1, tzinfo=timezone.utc)]
    calls = []

    def getter(k):
        calls.append(k)
        return f"v-{k}-{len(calls)}"

    cache = TTLCache(getter, lambda: t[0], timedelta(seconds=10))
    assert cache.get("a") == "v-a-1"
    assert cache.get("a") == "v-a-1"  # cached
    t[0] += timedelta(seconds=11)
    assert cache.get("a") == "v-a-2"  # expired, re-fetched
    assert calls == ["a", "a"]
