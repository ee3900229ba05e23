"""Visualize dataflow structure.

Parity target: ``bytewax.visualize`` (reference pysrc/bytewax/
visualize.py:30-405): render a `Dataflow` to the "rendered" data
model (ports resolved to globally-unique IDs, `from_port_ids`
pointing at each stream's originating port), encode it as JSON, or
draw a Mermaid flowchart; ``python -m bytewax_amd.visualize`` CLI.
The JSON schema and Mermaid text match the reference byte-for-byte
(its pytests/test_visualize.py runs green against this module).
"""

import json
from collections import ChainMap
from dataclasses import dataclass
from typing import Any, Dict, List

from .dataflow import Dataflow, Operator

__all__ = [
    "RenderedDataflow",
    "RenderedOperator",
    "RenderedPort",
    "render_dataflow",
    "to_rendered",
    "to_json",
    "to_mermaid",
]


@dataclass(frozen=True)
class RenderedPort:
    """Port with stream links resolved to originating ports."""

    port_name: str
    port_id: str
    from_port_ids: List[str]
    from_stream_ids: List[str]


@dataclass(frozen=True)
class RenderedOperator:
    """Operator with all ports resolved to globally-unique IDs."""

    op_type: str
    step_name: str
    step_id: str
    inp_ports: List[RenderedPort]
    out_ports: List[RenderedPort]
    substeps: List["RenderedOperator"]


@dataclass(frozen=True)
class RenderedDataflow:
    """Dataflow with streams, ports resolved to globally-unique IDs."""

    flow_id: str
    substeps: List[RenderedOperator]


def _port_name(port) -> str:
    return port.port_id.rsplit(".", 1)[1]


def _stream_ids(port) -> List[str]:
    if hasattr(port, "stream_id"):
        return [port.stream_id]
    return list(port.stream_ids.values())


def _to_rendered(step: Operator, stream_to_orig: ChainMap) -> RenderedOperator:
    inp_rports = [
        RenderedPort(
            _port_name(p),
            p.port_id,
            [stream_to_orig[s] for s in _stream_ids(p)],
            list(_stream_ids(p)),
        )
        for p in step.inp_ports
    ]
    stream_to_orig.update(
        {s: p.port_id for p in step.out_ports for s in _stream_ids(p)}
    )
    # Inner scope: a stream entering this step originates (for the
    # substeps) at this step's fake input port.
    inner = stream_to_orig.new_child(
        {s: p.port_id for p in step.inp_ports for s in _stream_ids(p)}
    )
    substeps = [_to_rendered(sub, inner) for sub in step.substeps]
    out_rports = [
        RenderedPort(
            _port_name(p),
            p.port_id,
            [inner[s] for s in _stream_ids(p)] if substeps else [],
            list(_stream_ids(p)) if substeps else [],
        )
        for p in step.out_ports
    ]
    return RenderedOperator(
        step.op_name,
        step.step_name,
        step.step_id,
        inp_rports,
        out_rports,
        substeps,
    )


def to_rendered(flow: Dataflow) -> RenderedDataflow:
    """Convert a dataflow into the "rendered" data model: all port
    links resolved, so a renderer only connects each
    `RenderedPort.port_id` to its `from_port_ids`."""
    stream_to_orig: ChainMap = ChainMap()
    return RenderedDataflow(
        flow.flow_id,
        [_to_rendered(s, stream_to_orig) for s in flow.substeps],
    )


#: Back-compat alias (pre-round-2 name).
render_dataflow = to_rendered


def _to_plain(obj: Any) -> Any:
    if isinstance(obj, (RenderedDataflow, RenderedOperator, RenderedPort)):
        return dict(
            typ=type(obj).__name__,
            **{
                f: _to_plain(getattr(obj, f))
                for f in obj.__dataclass_fields__
            },
        )
    if isinstance(obj, list):
        return [_to_plain(o) for o in obj]
    return obj


def to_json(flow: Dataflow) -> str:
    """Encode this dataflow into JSON (reference schema: every node
    carries a `typ` discriminator)."""
    return json.dumps(_to_plain(to_rendered(flow)), indent=2)


def to_mermaid(flow: Dataflow) -> str:
    """Render a dataflow as a Mermaid flowchart of the top-level
    steps, edges labeled `producer_port → consumer_port`."""
    rflow = to_rendered(flow)
    lines = ["flowchart TD", f'subgraph "{rflow.flow_id} (Dataflow)"']
    # Map every port id (at any depth) to its top-level step.
    port_step: Dict[str, RenderedOperator] = {}

    def index(top: RenderedOperator, step: RenderedOperator) -> None:
        for p in step.inp_ports + step.out_ports:
            port_step[p.port_id] = top
        for sub in step.substeps:
            index(top, sub)

    for step in rflow.substeps:
        index(step, step)
    port_names = {}

    def names(step: RenderedOperator) -> None:
        for p in step.inp_ports + step.out_ports:
            port_names[p.port_id] = p.port_name
        for sub in step.substeps:
            names(sub)

    for step in rflow.substeps:
        names(step)

    for step in rflow.substeps:
        lines.append(f'{step.step_id}["{step.step_name} ({step.op_type})"]')
        for p in step.inp_ports:
            for from_pid in p.from_port_ids:
                prod = port_step.get(from_pid)
                if prod is None or prod.step_id == step.step_id:
                    continue
                label = f"{port_names[from_pid]} → {p.port_name}"
                lines.append(
                    f'{prod.step_id} -- "{label}" --> {step.step_id}'
                )
    lines.append("end")
    return "\n".join(lines)


def _main() -> None:
    import argparse

    from .run import _locate_dataflow, _prepare_import

    parser = argparse.ArgumentParser(
        prog="python -m bytewax_amd.visualize",
        description="Render a dataflow's structure",
    )
    parser.add_argument("import_str")
    parser.add_argument(
        "--format", choices=["json", "mermaid"], default="mermaid"
    )
    args = parser.parse_args()
    flow = _locate_dataflow(*_prepare_import(args.import_str))
    if args.format == "json":
        print(to_json(flow))
    else:
        print(to_mermaid(flow))


if __name__ == "__main__":
    _main()
