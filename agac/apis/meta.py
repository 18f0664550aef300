"""Kubernetes-style object metadata and generic (de)serialization machinery.

Replaces k8s.io/apimachinery for this framework: every API object is a
dataclass with ``to_dict``/``from_dict`` producing the Kubernetes wire format
(camelCase JSON), so objects round-trip through the HTTP API server and
AdmissionReview payloads exactly like their Go counterparts.
"""

from __future__ import annotations

import copy
import dataclasses
import typing
from dataclasses import dataclass, field


def _camelize(name: str) -> str:
    parts = name.split("_")
    return parts[0] + "".join(p[:1].upper() + p[1:] for p in parts[1:])


def _is_api_type(tp) -> bool:
    return dataclasses.is_dataclass(tp) and isinstance(tp, type)


def _unwrap_optional(tp):
    origin = typing.get_origin(tp)
    if origin is typing.Union:
        args = [a for a in typing.get_args(tp) if a is not type(None)]
        if len(args) == 1:
            return args[0]
    return tp


def _serialize(value):
    if dataclasses.is_dataclass(value) and not isinstance(value, type):
        return to_dict(value)
    if isinstance(value, list):
        return [_serialize(v) for v in value]
    if isinstance(value, dict):
        return {k: _serialize(v) for k, v in value.items()}
    return value


def to_dict(obj) -> dict:
    """Serialize a dataclass API object to its camelCase wire format.

    ``None`` fields are omitted (Go ``omitempty`` behavior); empty lists and
    dicts are omitted too unless the field is listed in ``_keep_empty``.
    """
    overrides = getattr(type(obj), "_json_overrides", {})
    keep_empty = getattr(type(obj), "_keep_empty", set())
    out = {}
    for f in dataclasses.fields(obj):
        value = getattr(obj, f.name)
        if value is None:
            continue
        if value in ({}, []) and f.name not in keep_empty:
            continue
        json_name = overrides.get(f.name, _camelize(f.name))
        out[json_name] = _serialize(value)
    return out


def _deserialize(tp, value):
    tp = _unwrap_optional(tp)
    origin = typing.get_origin(tp)
    if origin is list:
        (item_tp,) = typing.get_args(tp)
        return [_deserialize(item_tp, v) for v in value or []]
    if origin is dict:
        return dict(value or {})
    if _is_api_type(tp):
        return from_dict(tp, value or {})
    return value


def from_dict(cls, data: dict):
    """Deserialize camelCase wire format into a dataclass API object.

    Unknown keys are ignored (forward compatibility, like Go json decoding).
    """
    overrides = getattr(cls, "_json_overrides", {})
    hints = typing.get_type_hints(cls)
    kwargs = {}
    for f in dataclasses.fields(cls):
        json_name = overrides.get(f.name, _camelize(f.name))
        if data is not None and json_name in data:
            kwargs[f.name] = _deserialize(hints[f.name], data[json_name])
    return cls(**kwargs)


@dataclass(slots=True)
class OwnerReference:
    """metav1.OwnerReference — modeled so controller writes against a real
    apiserver (full-replacement PUTs) never drop a user's owner links."""

    api_version: str = ""
    kind: str = ""
    name: str = ""
    uid: str = ""
    controller: typing.Optional[bool] = None
    block_owner_deletion: typing.Optional[bool] = None


@dataclass(slots=True)
class ObjectMeta:
    """Reference: metav1.ObjectMeta (the subset the controllers use, plus
    the write-safety fields for full-replacement updates)."""

    name: str = ""
    namespace: str = ""
    uid: str = ""
    resource_version: str = ""
    generation: int = 0
    creation_timestamp: typing.Optional[str] = None
    deletion_timestamp: typing.Optional[str] = None
    annotations: typing.Dict[str, str] = field(default_factory=dict)
    labels: typing.Dict[str, str] = field(default_factory=dict)
    finalizers: typing.List[str] = field(default_factory=list)
    owner_references: typing.List[OwnerReference] = field(default_factory=list)


@dataclass(slots=True)
class TypeMeta:
    kind: str = ""
    api_version: str = ""


_SCALARS = (str, int, float, bool, bytes, type(None))
_SCALAR_TYPES = {str, int, float, bool, bytes, type(None)}

# per-class compiled copier cache
_COPIER_CACHE: dict = {}


def _classify_hint(hint):
    """Map a field's type annotation to a copy strategy tag.
    Returns (tag, subclass|None); tag ∈ scalar / scalar_dict / scalar_list /
    dataclass / dataclass_list / fallback."""
    hint = _unwrap_optional(hint)
    if hint in _SCALAR_TYPES:
        return "scalar", None
    origin = typing.get_origin(hint)
    if origin is dict:
        args = typing.get_args(hint)
        if len(args) == 2 and args[1] in _SCALAR_TYPES:
            return "scalar_dict", None
        return "fallback", None
    if origin is list:
        args = typing.get_args(hint)
        if len(args) == 1:
            if args[0] in _SCALAR_TYPES:
                return "scalar_list", None
            if _is_api_type(args[0]):
                return "dataclass_list", args[0]
        return "fallback", None
    if _is_api_type(hint):
        return "dataclass", hint
    return "fallback", None


def _compile_copier(cls):
    """exec-generate a specialized copier for one dataclass: scalar fields
    assign directly, scalar dicts/lists shallow-copy (values immutable),
    nested API types recurse through their own compiled copier.  This is
    the framework's hottest path (the client-go generated-DeepCopy
    analogue): every store read/write and informer dispatch copies."""
    try:
        hints = typing.get_type_hints(cls)
    except Exception:
        hints = {}
    env = {"_cls": cls, "_new": object.__new__, "_deep": deep_copy}
    lines = ["def _copy(o):", "    n = _new(_cls)"]
    for f in dataclasses.fields(cls):
        name = f.name
        tag, sub = _classify_hint(hints.get(f.name, object))
        if tag == "scalar":
            lines.append(f"    n.{name} = o.{name}")
        elif tag == "scalar_dict":
            lines.append(f"    v = o.{name}; n.{name} = dict(v) if v is not None else None")
        elif tag == "scalar_list":
            lines.append(f"    v = o.{name}; n.{name} = list(v) if v is not None else None")
        elif tag == "dataclass":
            sub_copier = _copier_for(sub)
            env[f"_c_{name}"] = sub_copier
            lines.append(
                f"    v = o.{name}; n.{name} = _c_{name}(v) if v is not None else None"
            )
        elif tag == "dataclass_list":
            sub_copier = _copier_for(sub)
            env[f"_c_{name}"] = sub_copier
            lines.append(
                f"    v = o.{name}; "
                f"n.{name} = [_c_{name}(i) for i in v] if v is not None else None"
            )
        else:
            lines.append(f"    n.{name} = _deep(o.{name})")
    lines.append("    return n")
    exec("\n".join(lines), env)  # noqa: S102 - trusted, generated from dataclass fields
    return env["_copy"]


def _copier_for(cls):
    copier = _COPIER_CACHE.get(cls)
    if copier is None:
        # placeholder guards against recursive class graphs (none today)
        _COPIER_CACHE[cls] = lambda o: _generic_copy(o)
        copier = _compile_copier(cls)
        _COPIER_CACHE[cls] = copier
    return copier


def _generic_copy(obj):
    if isinstance(obj, _SCALARS):
        return obj
    if isinstance(obj, list):
        return [deep_copy(v) for v in obj]
    if isinstance(obj, dict):
        return {k: deep_copy(v) for k, v in obj.items()}
    if dataclasses.is_dataclass(obj):
        new = obj.__class__.__new__(obj.__class__)
        for f in dataclasses.fields(obj):
            setattr(new, f.name, deep_copy(getattr(obj, f.name)))
        return new
    return copy.deepcopy(obj)


def deep_copy(obj):
    """DeepCopy equivalent (reference uses generated DeepCopyObject).

    Per-class copiers are exec-compiled on first use from the dataclass's
    type annotations (scalars assigned, scalar containers shallow-copied,
    nested API types recursed) — the same codegen idea as the reference's
    zz_generated.deepcopy.go, done at runtime.  Fields holding values that
    don't match their annotation fall back to a generic recursive copy."""
    cls = obj.__class__
    copier = _COPIER_CACHE.get(cls)
    if copier is not None:
        return copier(obj)
    if isinstance(obj, _SCALARS):
        return obj
    if cls is list:
        return [deep_copy(v) for v in obj]
    if cls is dict:
        return {k: deep_copy(v) for k, v in obj.items()}
    if dataclasses.is_dataclass(obj) and not isinstance(obj, type):
        return _copier_for(cls)(obj)
    return copy.deepcopy(obj)


def meta_namespace_key(obj) -> str:
    """cache.MetaNamespaceKeyFunc: '<namespace>/<name>' or '<name>'."""
    meta = obj.metadata if hasattr(obj, "metadata") else obj
    if meta.namespace:
        return f"{meta.namespace}/{meta.name}"
    return meta.name


def split_meta_namespace_key(key: str):
    """cache.SplitMetaNamespaceKey: returns (namespace, name).

    Raises ValueError for keys with more than one '/'.
    """
    parts = key.split("/")
    if len(parts) == 1:
        return "", parts[0]
    if len(parts) == 2:
        return parts[0], parts[1]
    raise ValueError(f"unexpected key format: {key!r}")
