"""Streaming z-score anomaly detection per metric (reference
examples/anomaly_detector.py).

Two random metric streams merge; a per-key `stateful_map` keeps a
sliding window of the last 10 values and flags values more than two
standard deviations from the mean.
"""

import sys
from dataclasses import dataclass, field
from typing import List, Optional

from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import bytewax_amd.operators as op
from bytewax_amd.connectors.demo import RandomMetricSource
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow


@dataclass
class Detector:
    """Rolling mean / standard deviation over the last 10 samples."""

    window: List[float] = field(default_factory=list)
    mu: Optional[float] = None
    sigma: Optional[float] = None

    def observe(self, value: float) -> None:
        self.window.insert(0, value)
        del self.window[10:]
        n = len(self.window)
        self.mu = sum(self.window) / n
        var = sum((v - self.mu) ** 2 for v in self.window) / n
        self.sigma = var**0.5

    def is_anomalous(self, value: float, threshold_z: float) -> bool:
        if not self.mu or not self.sigma:
            return False
        return abs(value - self.mu) / self.sigma > threshold_z


def detect(state, value):
    state = state or Detector()
    flagged = state.is_anomalous(value, threshold_z=2.0)
    state.observe(value)
    return (state, (value, state.mu, state.sigma, flagged))


flow = Dataflow("anomaly_detector")
volts = op.input("inp_v", flow, RandomMetricSource("v_metric"))
hertz = op.input("inp_hz", flow, RandomMetricSource("hz_metric"))
metrics = op.merge("merge", volts, hertz)
labeled = op.stateful_map("detector", metrics, detect)


def fmt(key_value):
    metric, (value, mu, sigma, flagged) = key_value
    return (
        f"{metric}: value = {value}, mu = {mu:.2f}, "
        f"sigma = {sigma:.2f}, {flagged}"
    )


op.output("out", op.map("fmt", labeled, fmt), StdOutSink())

if __name__ == "__main__":
    from bytewax_amd.testing import run_main

    run_main(flow)
