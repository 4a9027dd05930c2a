"""Nested-structure pack/unpack (SURVEY.md §2 "Nested structure utils").

`nested_flatten` yields leaves of arbitrarily nested (tuple, list, dict)
structures in deterministic order; `nested_pack` rebuilds the structure from
a flat iterable; `nested_map` applies a function leaf-wise.
"""
from __future__ import annotations

from typing import Any, Iterable, Iterator


def nested_flatten(t: Any) -> Iterator[Any]:
    if isinstance(t, (list, tuple)):
        for item in t:
            yield from nested_flatten(item)
    elif isinstance(t, dict):
        for key in sorted(t.keys(), key=repr):
            yield from nested_flatten(t[key])
    else:
        yield t


def nested_pack(flat: Iterable[Any], structure: Any) -> Any:
    flat_iter = iter(flat)
    out = _nested_pack_inner(flat_iter, structure)
    try:
        next(flat_iter)
    except StopIteration:
        return out
    raise ValueError("nested_pack: too many leaves for structure")


def _nested_pack_inner(flat_iter: Iterator[Any], structure: Any) -> Any:
    if isinstance(structure, (list, tuple)):
        packed = [_nested_pack_inner(flat_iter, item) for item in structure]
        return type(structure)(packed)
    elif isinstance(structure, dict):
        keys = sorted(structure.keys(), key=repr)
        values = {k: _nested_pack_inner(flat_iter, structure[k]) for k in keys}
        return {k: values[k] for k in structure.keys()}
    else:
        try:
            return next(flat_iter)
        except StopIteration:
            raise ValueError("nested_pack: not enough leaves for structure") from None


def nested_map(fn, *structures: Any) -> Any:
    if not structures:
        raise ValueError("nested_map needs at least one structure")
    first = structures[0]
    flat_lists = [list(nested_flatten(s)) for s in structures]
    lengths = {len(f) for f in flat_lists}
    if len(lengths) != 1:
        raise ValueError(f"nested_map: structures have different leaf counts {lengths}")
    mapped = [fn(*leaves) for leaves in zip(*flat_lists)]
    return nested_pack(mapped, first)


def nested_compare(a: Any, b: Any) -> bool:
    """True if a and b have the same nesting structure (leaf values ignored)."""
    if isinstance(a, (list, tuple)):
        return (
            isinstance(b, type(a))
            and len(a) == len(b)
            and all(nested_compare(x, y) for x, y in zip(a, b))
        )
    if isinstance(a, dict):
        return (
            isinstance(b, dict)
            and a.keys() == b.keys()
            and all(nested_compare(a[k], b[k]) for k in a)
        )
    return not isinstance(b, (list, tuple, dict))
