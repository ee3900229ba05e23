"""Error types and utilities.

Parity target: ``bytewax.errors`` + the Rust error bridging
(reference src/errors.rs): runtime errors carry the step id of the
operator that raised.
"""

__all__ = ["BytewaxRuntimeError", "BytewaxTypeError"]


class BytewaxRuntimeError(RuntimeError):
    """An error occurred while executing a dataflow."""


class BytewaxTypeError(TypeError, RuntimeError):
    """A step's user callable violated an operator contract.

    Subclasses BOTH TypeError (the natural Python category) and
    RuntimeError: the reference surfaces these through its Rust
    panic bridging as RuntimeError from `run_main` (its pytests
    assert `raises(RuntimeError)`), so code written against either
    convention catches this."""
