"""Error types and utilities.

Parity target: ``bytewax.errors`` + the Rust error bridging
(reference src/errors.rs): runtime errors carry the step id of the
operator that raised.
"""

__all__ = ["BytewaxRuntimeError"]


class BytewaxRuntimeError(RuntimeError):
    """An error occurred while executing a dataflow."""
