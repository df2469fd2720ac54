"""Dataclass <-> camelCase-dict serde for the v1alpha2 API types.

The reference publishes its CRD schema as Go structs with camelCase JSON tags;
we keep byte-compatible YAML by converting snake_case dataclass fields to
camelCase on the wire. Nested dataclasses, lists and optionals round-trip.
"""
from __future__ import annotations

import copy
import dataclasses
import functools
import typing
from typing import Any, Dict, Type, TypeVar, get_args, get_origin

T = TypeVar("T")


@functools.lru_cache(maxsize=None)
def to_camel(name: str) -> str:
    head, *rest = name.split("_")
    return head + "".join(w.capitalize() for w in rest)


@functools.lru_cache(maxsize=None)
def _type_hints(cls):
    return typing.get_type_hints(cls)


@functools.lru_cache(maxsize=None)
def _camel_fields(cls):
    return {to_camel(f.name): f.name for f in dataclasses.fields(cls)}


def asdict(obj: Any, keep_none: bool = False) -> Any:
    """Serialize a dataclass tree to plain dicts with camelCase keys."""
    if dataclasses.is_dataclass(obj) and not isinstance(obj, type):
        out: Dict[str, Any] = {}
        for f in dataclasses.fields(obj):
            v = getattr(obj, f.name)
            if v is None and not keep_none:
                continue
            if v in ({}, []) and f.default_factory is not dataclasses.MISSING:  # type: ignore[misc]
                continue
            out[to_camel(f.name)] = asdict(v, keep_none)
        return out
    if isinstance(obj, dict):
        return {k: asdict(v, keep_none) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [asdict(v, keep_none) for v in obj]
    return obj


def _strip_optional(tp: Any) -> Any:
    if get_origin(tp) is typing.Union:
        args = [a for a in get_args(tp) if a is not type(None)]
        if len(args) == 1:
            return args[0]
    return tp


def fromdict(cls: Type[T], data: Any) -> T:
    """Deserialize camelCase dicts into the dataclass tree ``cls``."""
    if data is None:
        return None  # type: ignore[return-value]
    cls = _strip_optional(cls)
    origin = get_origin(cls)
    if origin in (list, tuple):
        (elem,) = get_args(cls) or (Any,)
        return [fromdict(elem, v) for v in data]  # type: ignore[return-value]
    if origin is dict:
        _, val_t = get_args(cls) or (Any, Any)
        return {k: fromdict(val_t, v) for k, v in data.items()}  # type: ignore[return-value]
    if dataclasses.is_dataclass(cls):
        if not isinstance(data, dict):
            raise TypeError(f"expected mapping for {cls.__name__}, got {type(data).__name__}")
        hints = _type_hints(cls)
        kwargs = {}
        known = _camel_fields(cls)
        for key, value in data.items():
            fname = known.get(key)
            if fname is None:
                continue  # forward-compatible: ignore unknown fields
            kwargs[fname] = fromdict(hints[fname], value)
        return cls(**kwargs)  # type: ignore[call-arg]
    return data  # scalar


def clone(obj: T) -> T:
    # dataclass trees of plain scalars/lists/dicts: deepcopy is ~10x the
    # dict round-trip and semantically identical here
    return copy.deepcopy(obj)
