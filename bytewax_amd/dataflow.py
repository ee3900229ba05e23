"""Dataflow graph builder.

This is the user-facing data model of the framework: a :class:`Dataflow`
is a tree of operator steps; leaves are one of the nine *core* operators
the engine knows how to execute, everything else is Python composition.

API parity target: ``bytewax.dataflow`` (reference ``pysrc/bytewax/
dataflow.py:178-715``) — ``Dataflow``, ``Stream`` (with ``then``),
the ``@operator`` decorator, and port types.  The implementation is
original: instead of generating a frozen dataclass per operator and
re-scoping stream ids through nested ports, we keep a scope *stack* per
flow and record the concrete argument payloads on each step.  The
engine only ever looks at core steps, whose payloads are explicit.
"""

import inspect
import re
from dataclasses import dataclass, field
from typing import (
    Any,
    Callable,
    Dict,
    Generic,
    Iterable,
    List,
    Optional,
    Tuple,
    TypeVar,
)

try:  # pragma: no cover - typing nicety only
    from typing import ParamSpec
except ImportError:  # pragma: no cover
    from typing_extensions import ParamSpec  # type: ignore

X = TypeVar("X")
X_co = TypeVar("X_co", covariant=True)
P = ParamSpec("P")
R = TypeVar("R")

__all__ = [
    "Dataflow",
    "DataflowId",
    "MultiPort",
    "Operator",
    "SinglePort",
    "Stream",
    "KeyedStream",
    "operator",
]

_STEP_ID_RE = re.compile(r"^[^.\s]+$")


@dataclass(frozen=True)
class DataflowId:
    """Unique ID of a dataflow."""

    flow_id: str


@dataclass(frozen=True)
class SinglePort:
    """A single input or output port on an operator step."""

    port_id: str
    stream_id: str


@dataclass(frozen=True)
class MultiPort:
    """A variadic input or output port on an operator step."""

    port_id: str
    stream_ids: Dict[str, str]


@dataclass
class Operator:
    """One step in a dataflow: either a core step or a composite.

    Core steps carry a ``payload`` dict the engine interprets (e.g. the
    mapper callable of ``flat_map_batch``).  Composite steps carry
    ``substeps``.
    """

    step_id: str  # fully-dotted path, e.g. "flow.count.stateful_batch"
    step_name: str  # last path segment
    op_name: str  # operator type, e.g. "flat_map_batch"
    is_core: bool
    inp_ports: List[Any] = field(default_factory=list)
    out_ports: List[Any] = field(default_factory=list)
    substeps: List["Operator"] = field(default_factory=list)
    payload: Dict[str, Any] = field(default_factory=dict)

    def __repr__(self) -> str:
        return f"<Operator {self.op_name} {self.step_id!r}>"


class _Scope:
    """Mutable builder state shared by all handles of one flow."""

    def __init__(self, flow: "Dataflow"):
        self.flow = flow
        # Stack of (prefix, substeps-list) the next step is appended to.
        self.stack: List[Tuple[str, List[Operator]]] = [
            (flow.flow_id, flow.substeps)
        ]
        self.seen_step_ids: set = set()

    @property
    def prefix(self) -> str:
        return self.stack[-1][0]

    @property
    def substeps(self) -> List[Operator]:
        return self.stack[-1][1]

    def push(self, step: Operator) -> None:
        self.stack.append((step.step_id, step.substeps))

    def pop(self) -> None:
        self.stack.pop()


@dataclass(frozen=True)
class Dataflow:
    """Dataflow definition.

    Use the operator functions in :mod:`bytewax_amd.operators` (e.g.
    ``op.input``) to add steps.
    """

    flow_id: str
    substeps: List[Operator] = field(default_factory=list)
    _scope: Optional[_Scope] = field(default=None, compare=False, repr=False)

    def __post_init__(self):
        if "." in self.flow_id:
            msg = "flow ID can't contain a period `.`"
            raise ValueError(msg)
        if self._scope is None:
            object.__setattr__(self, "_scope", _Scope(self))

    def flow(self) -> "Dataflow":
        return self


@dataclass(frozen=True)
class Stream(Generic[X_co]):
    """Handle to a stream of items you can add steps onto.

    You can reference the same stream multiple times to duplicate the
    data within.
    """

    stream_id: str
    _scope: _Scope = field(compare=False, repr=False)

    def flow(self) -> Dataflow:
        """The containing dataflow."""
        return self._scope.flow

    def then(
        self,
        op_fn: Callable[..., R],
        step_id: str,
        *args,
        **kwargs,
    ) -> R:
        """Chain a new step onto this stream.

        ``s.then(op.map, "x", f)`` is equivalent to
        ``op.map("x", s, f)``.
        """
        return op_fn(step_id, self, *args, **kwargs)


#: A stream of ``(key, value)`` 2-tuples.
KeyedStream = Stream


def _validate_step_id(step_id: str) -> None:
    if not isinstance(step_id, str):
        msg = f"step ID must be a `str`; got {step_id!r}"
        raise TypeError(msg)
    if not _STEP_ID_RE.match(step_id):
        msg = f"step ID {step_id!r} can't contain any periods or whitespace"
        raise ValueError(msg)


def _find_scope(args: Iterable[Any]) -> Optional[_Scope]:
    for a in args:
        if isinstance(a, (Stream, Dataflow)):
            return a._scope
        if isinstance(a, (list, tuple)):
            found = _find_scope(a)
            if found is not None:
                return found
    return None


def _ports_of(obj: Any, step_id: str, kind: str) -> List[Any]:
    """Derive port descriptors from an operator argument / return value."""
    ports: List[Any] = []
    idx = 0
    def visit(o: Any) -> None:
        nonlocal idx
        if isinstance(o, Stream):
            ports.append(SinglePort(f"{step_id}.{kind}{idx}", o.stream_id))
            idx += 1
        elif isinstance(o, Dataflow):
            pass
        elif isinstance(o, (list, tuple)):
            for el in o:
                visit(el)
        elif hasattr(o, "__dataclass_fields__"):
            for f in o.__dataclass_fields__:
                visit(getattr(o, f))

    visit(obj)
    return ports


def _named_ports(named_values, step_id: str) -> List[Any]:
    """Port descriptors named like the reference: input ports carry
    the builder's parameter names ("up", "lefts", ...), output ports
    the returned port names ("down") or dataclass field names
    ("trues"/"falses"/...)."""
    ports: List[Any] = []
    for name, o in named_values:
        if isinstance(o, Stream):
            ports.append(SinglePort(f"{step_id}.{name}", o.stream_id))
        elif isinstance(o, (list, tuple)):
            if all(isinstance(el, Stream) for el in o) and o:
                ports.append(
                    MultiPort(
                        f"{step_id}.{name}",
                        {str(i): el.stream_id for i, el in enumerate(o)},
                    )
                )
            else:
                for i, el in enumerate(o):
                    ports.extend(
                        _named_ports([(f"{name}_{i}", el)], step_id)
                    )
        elif isinstance(o, Dataflow) or o is None:
            pass
        elif hasattr(o, "__dataclass_fields__"):
            ports.extend(
                _named_ports(
                    [
                        (f, getattr(o, f))
                        for f in o.__dataclass_fields__
                    ],
                    step_id,
                )
            )
    return ports


def operator(builder: Optional[Callable] = None, *, _core: bool = False):
    """Decorate a function to make it a dataflow operator.

    The decorated function's first parameter must be ``step_id`` and at
    least one other parameter must be a :class:`Stream` or
    :class:`Dataflow`.  Calling it appends a step to the flow; the body
    (for non-core operators) builds substeps which are nested under the
    new step's scope.

    Reference parity: ``bytewax.dataflow.operator``
    (dataflow.py:697-715).
    """

    def decorate(fn: Callable) -> Callable:
        sig = inspect.signature(fn)
        params = list(sig.parameters)
        if not params or params[0] != "step_id":
            msg = (
                f"operator builder {fn.__name__!r} must take `step_id` "
                "as its first parameter"
            )
            raise TypeError(msg)
        op_name = fn.__name__

        def wrapper(step_id: str, *args, **kwargs):
            _validate_step_id(step_id)
            scope = _find_scope(args) or _find_scope(kwargs.values())
            if scope is None:
                msg = (
                    f"operator {op_name!r} requires at least one "
                    "upstream argument that must be a `Stream` (or a "
                    "`Dataflow`) to anchor it to a flow"
                )
                raise TypeError(msg)
            full_id = f"{scope.prefix}.{step_id}"
            if full_id in scope.seen_step_ids:
                msg = f"duplicate step ID {full_id!r}"
                raise ValueError(msg)
            scope.seen_step_ids.add(full_id)

            step = Operator(
                step_id=full_id,
                step_name=step_id,
                op_name=op_name,
                is_core=_core,
            )
            _b = sig.bind(full_id, *args, **kwargs)
            step.inp_ports = _named_ports(
                [
                    (n, v)
                    for n, v in _b.arguments.items()
                    if n != "step_id"
                ],
                full_id,
            )
            scope.substeps.append(step)
            # Builders see the fully-qualified step id (matching the
            # reference: `op.inspect("help", s)` prints "flow.help").
            if _core:
                # Core ops don't have substeps; record the payload by
                # binding the builder signature (the builder body of a
                # core op is only validation).
                bound = sig.bind(full_id, *args, **kwargs)
                bound.apply_defaults()
                step.payload = dict(bound.arguments)
                out = fn(full_id, *args, **kwargs)
                # The core builder returns stream *names*: turn them
                # into concrete Streams scoped to this step.
                out = _materialize_out(out, step, scope)
            else:
                scope.push(step)
                try:
                    out = fn(full_id, *args, **kwargs)
                finally:
                    scope.pop()
            if isinstance(out, Stream):
                step.out_ports = _named_ports([("down", out)], full_id)
            else:
                step.out_ports = _named_ports([("out", out)], full_id)
            return out

        wrapper.__name__ = op_name
        wrapper.__qualname__ = getattr(fn, "__qualname__", op_name)
        wrapper.__doc__ = fn.__doc__
        wrapper.__module__ = fn.__module__
        wrapper.__wrapped__ = fn
        wrapper._is_operator = True
        wrapper._is_core = _core
        return wrapper

    if builder is not None:
        return decorate(builder)
    return decorate


def _materialize_out(out: Any, step: Operator, scope: _Scope) -> Any:
    """Convert port-name declarations returned by a core builder into
    Streams whose ids live under the step id."""
    if out is None:
        return None
    if isinstance(out, str):
        return Stream(f"{step.step_id}.{out}", scope)
    if isinstance(out, tuple):
        return tuple(_materialize_out(o, step, scope) for o in out)
    if hasattr(out, "__dataclass_fields__"):
        import dataclasses as _dc

        repl = {
            f: _materialize_out(getattr(out, f), step, scope)
            for f in out.__dataclass_fields__
        }
        return _dc.replace(out, **repl)
    return out
