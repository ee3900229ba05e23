"""Level-2 order book maintenance per product (reference
examples/orderbook.py).

The reference streams Coinbase L2 updates over a websocket; this
environment has no network, so a deterministic synthetic feed of the
same message shape (snapshot + l2update changes) drives the identical
book-maintenance dataflow: per-product `stateful_map` holding the
bid/ask depth maps, emitting best-bid/best-ask/spread summaries and
filtering for wide spreads.
"""

import random
import sys
from dataclasses import dataclass, field
from pathlib import Path
from typing import Dict, List, Optional

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.inputs import FixedPartitionedSource, StatefulSourcePartition

PRODUCTS = ["BTC-USD", "ETH-USD"]
MID = {"BTC-USD": 64_000.0, "ETH-USD": 3_000.0}
N_UPDATES = 40


class SyntheticL2Partition(StatefulSourcePartition):
    """Replayable feed: one snapshot, then random depth changes."""

    def __init__(self, product: str, resume_i: Optional[int]):
        self.product = product
        self.i = resume_i or 0
        self.rng = random.Random(sum(product.encode()))
        mid = MID[product]
        self._msgs = []
        bids = [[f"{mid - j - 1:.2f}", f"{self.rng.random():.4f}"] for j in range(5)]
        asks = [[f"{mid + j + 1:.2f}", f"{self.rng.random():.4f}"] for j in range(5)]
        self._msgs.append(
            (product, {"type": "snapshot", "bids": bids, "asks": asks})
        )
        for _ in range(N_UPDATES):
            side = self.rng.choice(["buy", "sell"])
            off = self.rng.randint(1, 6)
            price = mid - off if side == "buy" else mid + off
            size = self.rng.choice([0.0, self.rng.random()])
            self._msgs.append(
                (
                    product,
                    {
                        "type": "l2update",
                        "changes": [[side, f"{price:.2f}", f"{size:.4f}"]],
                    },
                )
            )

    def next_batch(self):
        if self.i >= len(self._msgs):
            raise StopIteration()
        msg = self._msgs[self.i]
        self.i += 1
        return [msg]

    def snapshot(self):
        return self.i


@dataclass
class SyntheticL2Source(FixedPartitionedSource):
    product_ids: List[str]

    def list_parts(self):
        return self.product_ids

    def build_part(self, step_id, for_part, resume_state):
        return SyntheticL2Partition(for_part, resume_state)


@dataclass(frozen=True)
class Summary:
    bid_price: float
    bid_size: float
    ask_price: float
    ask_size: float
    spread: float


@dataclass
class OrderBook:
    bids: Dict[float, float] = field(default_factory=dict)
    asks: Dict[float, float] = field(default_factory=dict)

    def apply(self, msg) -> None:
        if not self.bids and "bids" in msg:
            self.bids = {float(p): float(s) for p, s in msg["bids"]}
        if not self.asks and "asks" in msg:
            self.asks = {float(p): float(s) for p, s in msg["asks"]}
        for side, price_s, size_s in msg.get("changes", []):
            price, size = float(price_s), float(size_s)
            book = self.bids if side == "buy" else self.asks
            if size == 0.0:
                book.pop(price, None)
            else:
                book[price] = size

    def summarize(self) -> Optional[Summary]:
        if not self.bids or not self.asks:
            return None  # one side fully consumed; no quote
        bid = max(self.bids)
        ask = min(self.asks)
        return Summary(bid, self.bids[bid], ask, self.asks[ask], ask - bid)


def maintain(state, msg):
    state = state or OrderBook()
    state.apply(msg)
    return (state, state.summarize())


flow = Dataflow("orderbook")
feed = op.input("inp", flow, SyntheticL2Source(PRODUCTS))
books = op.stateful_map("book", feed, maintain)


def wide_spread(prod_summary):
    _product, s = prod_summary
    return s is not None and s.spread / s.ask_price > 0.0001


wide = op.filter("wide_spread", books, wide_spread)
op.output("out", wide, StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)
