"""Small generic utility functions."""

from typing import Callable, Iterable, List, Tuple, TypeVar

X = TypeVar("X")

__all__ = ["partition"]


def partition(
    it: Iterable[X], predicate: Callable[[X], bool]
) -> Tuple[List[X], List[X]]:
    """Split an iterable in two based on a predicate.

    :returns: (items where predicate true, items where false).
    """
    trues: List[X] = []
    falses: List[X] = []
    for x in it:
        if predicate(x):
            trues.append(x)
        else:
            falses.append(x)
    return (trues, falses)
