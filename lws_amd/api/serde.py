"""Dataclass <-> JSON-dict serde with Kubernetes-style camelCase keys.

The whole lws_amd API layer is plain Python dataclasses; this module gives
them a stable wire format (camelCase, omit-empty) so that revision hashing,
deep copies, strategic patches and the YAML/JSON client all share one
canonical representation.  Mirrors the role the generated deepcopy/JSON
machinery plays in the reference (zz_generated.deepcopy.go,
client-go applyconfigurations) without code generation.
"""
from __future__ import annotations

import dataclasses
import typing
from typing import Any, Optional, Union, get_args, get_origin, get_type_hints


def snake_to_camel(name: str) -> str:
    parts = name.split("_")
    return parts[0] + "".join(p.title() for p in parts[1:])


_HINTS_CACHE: dict[type, dict[str, Any]] = {}


def _hints(cls: type) -> dict[str, Any]:
    h = _HINTS_CACHE.get(cls)
    if h is None:
        h = get_type_hints(cls)
        _HINTS_CACHE[cls] = h
    return h


def _is_optional(tp: Any) -> bool:
    return get_origin(tp) is Union and type(None) in get_args(tp)


def _strip_optional(tp: Any) -> Any:
    if get_origin(tp) is Union:
        args = [a for a in get_args(tp) if a is not type(None)]
        if len(args) == 1:
            return args[0]
        return Union[tuple(args)]
    return tp


def to_dict(obj: Any, *, omit_empty: bool = True) -> Any:
    """Recursively serialize a dataclass to a JSON-compatible dict.

    None values are always dropped; empty lists/dicts are dropped when
    omit_empty (matching k8s `omitempty` semantics so that hashes of
    semantically-equal objects are equal).
    """
    if dataclasses.is_dataclass(obj) and not isinstance(obj, type):
        out = {}
        for f in dataclasses.fields(obj):
            v = getattr(obj, f.name)
            if v is None:
                continue
            sv = to_dict(v, omit_empty=omit_empty)
            if omit_empty and (sv == [] or sv == {}):
                continue
            key = f.metadata.get("json", snake_to_camel(f.name))
            out[key] = sv
        return out
    if isinstance(obj, dict):
        return {k: to_dict(v, omit_empty=omit_empty) for k, v in obj.items() if v is not None}
    if isinstance(obj, (list, tuple)):
        return [to_dict(v, omit_empty=omit_empty) for v in obj]
    if isinstance(obj, (str, int, float, bool)) or obj is None:
        return obj
    return str(obj)


def from_dict(cls: Any, data: Any) -> Any:
    """Recursively build a dataclass from a JSON-compatible dict."""
    if data is None:
        return None
    cls = _strip_optional(cls)
    origin = get_origin(cls)
    if origin in (list, tuple):
        (item_tp,) = get_args(cls) or (Any,)
        return [from_dict(item_tp, v) for v in data]
    if origin is dict:
        args = get_args(cls)
        vt = args[1] if len(args) == 2 else Any
        return {k: from_dict(vt, v) for k, v in data.items()}
    if origin is Union:  # e.g. IntOrString = Union[int, str]
        return data
    if dataclasses.is_dataclass(cls):
        hints = _hints(cls)
        kwargs = {}
        by_json_key = {}
        for f in dataclasses.fields(cls):
            by_json_key[f.metadata.get("json", snake_to_camel(f.name))] = f
        for key, v in (data or {}).items():
            f = by_json_key.get(key)
            if f is None:
                continue  # tolerate unknown fields like the k8s API server
            kwargs[f.name] = from_dict(hints[f.name], v)
        return cls(**kwargs)
    if cls in (Any, object) or isinstance(cls, typing.TypeVar):
        return data
    if cls is float and isinstance(data, int):
        return float(data)
    return data


def deep_copy(obj: Any) -> Any:
    """Deep copy (DeepCopyObject equivalent).  Uses pickle's C path — ~10x
    faster than a serialize/deserialize round trip through the JSON form,
    which matters because the store copies on every read/write/dispatch."""
    if obj is None:
        return None
    import pickle

    return pickle.loads(pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL))


def jfield(json_key: str, **kw: Any) -> Any:
    """dataclasses.field with an explicit JSON key override."""
    return dataclasses.field(metadata={"json": json_key}, **kw)
