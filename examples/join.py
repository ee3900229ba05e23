"""Multi-sided keyed join (reference examples/join.py).

Four value streams derived from three sources join on user id; the
joined tuple emits once every side has a value ("complete" mode).
"""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.testing import TestingSource

flow = Dataflow("join")


def by_user(rec):
    return str(rec["user_id"])


profiles = op.input(
    "profiles", flow, TestingSource([{"user_id": 123, "name": "Bumble"}])
)
names = op.map_value(
    "name", op.key_on("k1", profiles, by_user), lambda r: r["name"]
)

contacts = op.input(
    "contacts",
    flow,
    TestingSource([{"user_id": 123, "email": "bee@example.com"}]),
)
emails = op.map_value(
    "email", op.key_on("k2", contacts, by_user), lambda r: r["email"]
)

prefs = op.input(
    "prefs",
    flow,
    TestingSource([{"user_id": 123, "color": "yellow", "sound": "buzz"}]),
)
keyed_prefs = op.key_on("k3", prefs, by_user)
colors = op.map_value("color", keyed_prefs, lambda r: r["color"])
sounds = op.map_value("sound", keyed_prefs, lambda r: r["sound"])

joined = op.join("join", names, emails, colors, sounds)
op.output("out", joined, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)
