"""Helper functions for using operators.

Parity target: ``bytewax.operators.helpers`` (reference
operators/helpers.py).
"""

from typing import Callable, Dict, TypeVar

K = TypeVar("K")
V = TypeVar("V")

__all__ = ["map_dict_value"]


def map_dict_value(
    key: K, mapper: Callable[[V], V]
) -> Callable[[Dict[K, V]], Dict[K, V]]:
    """Build a mapper that transforms one value in a dict in place and
    returns the dict (a simple lens for the `map` operator).

    :arg key: Dictionary key.
    :arg mapper: Function to run on the value for that key.
    :returns: A function performing that mapping when called.
    """

    def shim_mapper(obj: Dict[K, V]) -> Dict[K, V]:
        obj[key] = mapper(obj[key])
        return obj

    return shim_mapper
