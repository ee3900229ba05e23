"""Flatten a Dataflow's operator tree into an executable core graph.

Mirrors the role of the reference's dataflow compiler
(reference src/worker.rs:255-497) which stack-walks the Python operator
tree and instantiates one Timely operator per core step; here we
produce a flat list of :class:`CoreStep` plus consumer wiring and
exchange metadata for the BSP scheduler.
"""

from dataclasses import dataclass, field
from typing import Any, Dict, List, Tuple

from ..dataflow import Dataflow, Operator, Stream

CORE_OPS = {
    "_noop",
    "branch",
    "flat_map_batch",
    "input",
    "inspect_debug",
    "merge",
    "output",
    "redistribute",
    "stateful_batch",
}

# Output port names per core op, in order.
_OUT_PORTS = {
    "_noop": ["down"],
    "branch": ["trues", "falses"],
    "flat_map_batch": ["down"],
    "input": ["down"],
    "inspect_debug": ["down"],
    "merge": ["down"],
    "output": [],
    "redistribute": ["down"],
    "stateful_batch": ["down"],
}


@dataclass
class CoreStep:
    idx: int
    step_id: str
    op_name: str
    payload: Dict[str, Any]
    inp_streams: List[str]
    out_streams: List[str]
    # How items must be routed to this step across workers:
    # "local" | "key" | "random" | "part"
    exchange: str = "local"


@dataclass
class ExecGraph:
    flow_id: str
    steps: List[CoreStep]
    # stream_id -> [(consumer step idx, consumer input idx)]
    consumers: Dict[str, List[Tuple[int, int]]] = field(default_factory=dict)
    # number of exchange-delivery rounds needed per scheduling pass
    n_exchange_rounds: int = 0
    # step idxs in topological order
    topo: List[int] = field(default_factory=list)


def _collect_core(steps: List[Operator], out: List[Operator]) -> None:
    for s in steps:
        if s.is_core:
            out.append(s)
        else:
            _collect_core(s.substeps, out)


def _inp_streams(op: Operator) -> List[str]:
    payload = op.payload
    if op.op_name == "input":
        return []
    if op.op_name == "merge":
        return [s.stream_id for s in payload["ups"]]
    up = payload.get("up")
    if isinstance(up, Stream):
        return [up.stream_id]
    msg = f"core step {op.step_id!r} has no upstream recorded"
    raise AssertionError(msg)


def compile_graph(flow: Dataflow) -> ExecGraph:
    """Flatten the operator tree and wire streams to consumers."""
    from ..outputs import FixedPartitionedSink

    core: List[Operator] = []
    _collect_core(flow.substeps, core)

    steps: List[CoreStep] = []
    for i, op in enumerate(core):
        out_streams = [
            f"{op.step_id}.{port}" for port in _OUT_PORTS[op.op_name]
        ]
        exchange = "local"
        if op.op_name == "stateful_batch":
            exchange = "key"
        elif op.op_name == "redistribute":
            exchange = "random"
        elif op.op_name == "output" and isinstance(
            op.payload.get("sink"), FixedPartitionedSink
        ):
            exchange = "part"
        steps.append(
            CoreStep(
                idx=i,
                step_id=op.step_id,
                op_name=op.op_name,
                payload=op.payload,
                inp_streams=_inp_streams(op),
                out_streams=out_streams,
                exchange=exchange,
            )
        )

    graph = ExecGraph(flow_id=flow.flow_id, steps=steps)

    producers: Dict[str, int] = {}
    for s in steps:
        for sid in s.out_streams:
            producers[sid] = s.idx
    n_inputs = 0
    n_outputs = 0
    for s in steps:
        if s.op_name == "input":
            n_inputs += 1
        if s.op_name in ("output", "inspect_debug"):
            # `inspect` counts as a terminal step like the reference
            # (its inspect-only pytest flows run without an
            # `op.output`).
            n_outputs += 1
        for inp_idx, sid in enumerate(s.inp_streams):
            if sid not in producers:
                msg = (
                    f"stream {sid!r} consumed by {s.step_id!r} is not "
                    "produced by any step"
                )
                raise AssertionError(msg)
            graph.consumers.setdefault(sid, []).append((s.idx, inp_idx))

    # Missing input/output surface as RuntimeError like the
    # reference (its panic bridging; pytests assert RuntimeError and
    # NOT ValueError).
    from ..errors import BytewaxRuntimeError

    if n_inputs < 1:
        msg = "Dataflow needs to contain at least one input step"
        raise BytewaxRuntimeError(msg)
    if n_outputs < 1:
        msg = "Dataflow needs to contain at least one output step"
        raise BytewaxRuntimeError(msg)

    # Topological order + exchange depth (number of exchange edges on
    # the longest path ending at each step).
    indeg = {s.idx: 0 for s in steps}
    adj: Dict[int, List[int]] = {s.idx: [] for s in steps}
    for s in steps:
        for sid in s.inp_streams:
            p = producers[sid]
            adj[p].append(s.idx)
            indeg[s.idx] += 1
    ready = [i for i, d in indeg.items() if d == 0]
    topo: List[int] = []
    depth = {i: 0 for i in indeg}
    while ready:
        i = ready.pop()
        topo.append(i)
        for j in adj[i]:
            d = depth[i] + (1 if steps[j].exchange != "local" else 0)
            depth[j] = max(depth[j], d)
            indeg[j] -= 1
            if indeg[j] == 0:
                ready.append(j)
    if len(topo) != len(steps):
        msg = "dataflow graph contains a cycle"
        raise ValueError(msg)
    graph.topo = topo
    graph.n_exchange_rounds = max(depth.values()) if depth else 0
    return graph
