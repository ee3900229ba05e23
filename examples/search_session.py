"""Sessionize per-user search activity and compute click-through rate
(reference examples/search_session.py).

Events are keyed by user, collected into gap-based session windows,
split into per-search sub-sessions, and reduced to a CTR per search.
"""

import sys
from dataclasses import dataclass
from datetime import datetime, timedelta, timezone
from pathlib import Path
from typing import List

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
import bytewax_amd.operators.windowing as win
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.operators.windowing import EventClock, SessionWindower
from bytewax_amd.testing import TestingSource


@dataclass
class Event:
    user: int
    dt: datetime


@dataclass
class AppOpen(Event): ...


@dataclass
class Search(Event):
    query: str = ""


@dataclass
class Results(Event):
    items: List[str] = None


@dataclass
class ClickResult(Event):
    item: str = ""


@dataclass
class AppClose(Event): ...


START = datetime(2024, 1, 1, tzinfo=timezone.utc)


def at(seconds: int) -> datetime:
    return START + timedelta(seconds=seconds)


CLIENT_EVENTS = [
    AppOpen(user=1, dt=START),
    Search(user=1, dt=at(1), query="dogs"),
    Results(user=1, dt=at(2), items=["fido", "rover", "buddy"]),
    ClickResult(user=1, dt=at(3), item="rover"),
    Search(user=1, dt=at(4), query="cats"),
    Results(user=1, dt=at(5), items=["fluffy", "burrito", "kathy"]),
    ClickResult(user=1, dt=at(6), item="fluffy"),
    AppOpen(user=2, dt=at(7)),
    ClickResult(user=1, dt=at(8), item="kathy"),
    Search(user=2, dt=at(9), query="fruit"),
    AppClose(user=1, dt=at(10)),
    AppClose(user=2, dt=at(11)),
]


def split_searches(session: List[Event]):
    """Split one user session into per-search sub-sessions."""
    current: List[Event] = []
    for ev in session:
        if isinstance(ev, Search):
            if current:
                yield current
            current = [ev]
        elif current:
            current.append(ev)
    if current:
        yield current


def ctr(search_session: List[Event]) -> float:
    clicks = sum(1 for ev in search_session if isinstance(ev, ClickResult))
    searches = sum(1 for ev in search_session if isinstance(ev, Search))
    return clicks / searches if searches else 0.0


flow = Dataflow("search_session")
events = op.input("inp", flow, TestingSource(CLIENT_EVENTS))
keyed = op.key_on("user", events, lambda ev: str(ev.user))
clock = EventClock(
    ts_getter=lambda ev: ev.dt, wait_for_system_duration=timedelta(0)
)
sessions = win.collect_window(
    "sessions", keyed, clock, SessionWindower(gap=timedelta(seconds=5))
)
per_search = op.flat_map_value(
    "split", sessions.down, lambda wid_vals: list(split_searches(wid_vals[1]))
)
rates = op.map_value("ctr", per_search, ctr)
op.output("out", rates, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)
