"""Visualize dataflow structure.

Parity target: ``bytewax.visualize`` (reference pysrc/bytewax/
visualize.py:52-405): render a `Dataflow` to a JSON-able structure or
Mermaid diagram; ``python -m bytewax_amd.visualize`` CLI.
"""

import json
from dataclasses import dataclass
from typing import Any, Dict, List

from .dataflow import Dataflow, Operator

__all__ = ["RenderedDataflow", "RenderedOperator", "render_dataflow", "to_json", "to_mermaid"]


@dataclass(frozen=True)
class RenderedPort:
    port_id: str
    stream_ids: List[str]


@dataclass(frozen=True)
class RenderedOperator:
    op_name: str
    step_id: str
    inp_ports: List[RenderedPort]
    out_ports: List[RenderedPort]
    substeps: List["RenderedOperator"]


@dataclass(frozen=True)
class RenderedDataflow:
    flow_id: str
    substeps: List[RenderedOperator]


def _render_op(op: Operator) -> RenderedOperator:
    def ports(ps):
        out = []
        for p in ps:
            if hasattr(p, "stream_id"):
                out.append(RenderedPort(p.port_id, [p.stream_id]))
            elif hasattr(p, "stream_ids"):
                out.append(RenderedPort(p.port_id, list(p.stream_ids.values())))
        return out

    return RenderedOperator(
        op.op_name,
        op.step_id,
        ports(op.inp_ports),
        ports(op.out_ports),
        [_render_op(s) for s in op.substeps],
    )


def render_dataflow(flow: Dataflow) -> RenderedDataflow:
    """Convert a dataflow into a renderable structure."""
    return RenderedDataflow(
        flow.flow_id, [_render_op(s) for s in flow.substeps]
    )


def _to_plain(obj: Any) -> Any:
    if hasattr(obj, "__dataclass_fields__"):
        return {
            f: _to_plain(getattr(obj, f)) for f in obj.__dataclass_fields__
        }
    if isinstance(obj, list):
        return [_to_plain(o) for o in obj]
    return obj


def to_json(flow: Dataflow) -> str:
    """Encode this dataflow into JSON."""
    return json.dumps(_to_plain(render_dataflow(flow)), indent=2)


def to_mermaid(flow: Dataflow) -> str:
    """Render a dataflow as a Mermaid flowchart (top-level steps)."""
    lines = ["flowchart TD", f'subgraph "{flow.flow_id} (Dataflow)"']
    stream_producers: Dict[str, str] = {}

    def walk_core(op: Operator):
        if op.is_core:
            yield op
        for s in op.substeps:
            yield from walk_core(s)

    top = list(flow.substeps)
    for op in top:
        lines.append(f'{op.step_id}["{op.step_name} ({op.op_name})"]')
        for core in walk_core(op):
            for sid in [f"{core.step_id}.{p}" for p in ("down", "trues", "falses")]:
                stream_producers[sid] = op.step_id
    for op in top:
        seen = set()
        for core in walk_core(op):
            for p in core.inp_ports:
                for sid in (
                    [p.stream_id]
                    if hasattr(p, "stream_id")
                    else list(p.stream_ids.values())
                ):
                    prod = stream_producers.get(sid)
                    if prod and prod != op.step_id and (prod, op.step_id) not in seen:
                        seen.add((prod, op.step_id))
                        lines.append(f"{prod} --> {op.step_id}")
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
