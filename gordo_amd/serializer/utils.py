"""Typing helpers for signature-driven coercion
(spec: gordo/serializer/utils.py:16-48)."""
from __future__ import annotations

import typing


def is_tuple_type(annotation) -> bool:
    """
    True when a type annotation denotes a tuple.

    >>> import typing
    >>> is_tuple_type(typing.Tuple[int, int])
    True
    >>> is_tuple_type(tuple)
    True
    >>> is_tuple_type(typing.List[int])
    False
    """
    if annotation is tuple:
        return True
    origin = typing.get_origin(annotation)
    if origin is tuple:
        return True
    if origin is typing.Union:
        return any(is_tuple_type(a) for a in typing.get_args(annotation))
    return False
