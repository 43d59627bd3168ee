"""Call-signature inference & validation.

Reference capability (pylzy/lzy/core/call.py:271-334 +
api/v1/utils/types.py): bind args to the function signature, infer output
types from the return annotation (a ``Tuple[...]`` annotation means
multiple outputs), and check declared vs actual argument types.
"""
from __future__ import annotations

import inspect
import typing
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, Optional, Sequence, Tuple, Type, get_type_hints


@dataclass
class CallSignature:
    func: Callable
    args: Tuple[Any, ...]
    kwargs: Dict[str, Any]
    arg_names: Tuple[str, ...]
    arg_types: Tuple[Type, ...]
    kwarg_types: Dict[str, Type]

    @property
    def qualname(self) -> str:
        return getattr(self.func, "__qualname__", getattr(self.func, "__name__", "op"))


class _Absence:
    pass


ABSENT = _Absence()


def _unwrap_annotation(ann: Any) -> Type:
    origin = typing.get_origin(ann)
    if origin is typing.Union:  # Optional[T] -> T (first non-None)
        args = [a for a in typing.get_args(ann) if a is not type(None)]
        return _unwrap_annotation(args[0]) if args else type(None)
    if origin is not None:
        return origin  # List[int] -> list, etc.
    if ann is None:
        return type(None)
    if isinstance(ann, type):
        return ann
    return object


def infer_return_types(func: Callable) -> Optional[Sequence[Type]]:
    """None if the return annotation is missing (caller raises TypeError —
    reference op.py:36-44 behavior)."""
    try:
        hints = get_type_hints(func)
    except Exception:
        hints = getattr(func, "__annotations__", {})
    if "return" not in hints:
        return None
    ann = hints["return"]
    if ann is None or ann is type(None):
        return (type(None),)
    origin = typing.get_origin(ann)
    if origin is tuple:
        args = typing.get_args(ann)
        if args and args[-1] is not Ellipsis:
            return tuple(_unwrap_annotation(a) for a in args)
    return (_unwrap_annotation(ann),)


_SIG_CACHE: dict = {}


def _sig_and_hints(func: Callable):
    """inspect.signature + get_type_hints are pure per function —
    memoized (they cost ~0.1 ms per lazy call otherwise)."""
    key = id(func)
    hit = _SIG_CACHE.get(key)
    if hit is not None and hit[0] is func:  # id() reuse guard
        return hit[1], hit[2]
    sig = inspect.signature(func)
    try:
        hints = get_type_hints(func)
    except Exception:
        hints = getattr(func, "__annotations__", {})
    _SIG_CACHE[key] = (func, sig, hints)
    return sig, hints


def infer_call_signature(func: Callable, *args: Any, **kwargs: Any) -> CallSignature:
    sig, hints = _sig_and_hints(func)
    bound = sig.bind(*args, **kwargs)
    bound.apply_defaults()

    arg_names = []
    arg_types = []
    pos_args = []
    kw = {}
    kw_types = {}
    for name, param in sig.parameters.items():
        if name not in bound.arguments:
            continue
        value = bound.arguments[name]
        declared = _unwrap_annotation(hints.get(name, object))
        if param.kind in (
            inspect.Parameter.POSITIONAL_ONLY,
            inspect.Parameter.POSITIONAL_OR_KEYWORD,
        ):
            arg_names.append(name)
            arg_types.append(declared)
            pos_args.append(value)
        elif param.kind is inspect.Parameter.VAR_POSITIONAL:
            for i, v in enumerate(value):
                arg_names.append(f"{name}_{i}")
                arg_types.append(declared)
                pos_args.append(v)
        elif param.kind is inspect.Parameter.KEYWORD_ONLY:
            kw[name] = value
            kw_types[name] = declared
        elif param.kind is inspect.Parameter.VAR_KEYWORD:
            for k, v in value.items():
                kw[k] = v
                kw_types[k] = object

    return CallSignature(
        func=func,
        args=tuple(pos_args),
        kwargs=kw,
        arg_names=tuple(arg_names),
        arg_types=tuple(arg_types),
        kwarg_types=kw_types,
    )


def check_type_compatible(value: Any, declared: Type) -> bool:
    """Lenient declared-vs-actual check (reference call.py:306-312): used
    to warn, not to block — lazy proxies and duck typing stay usable."""
    if declared is object:
        return True
    if value is None:
        # Optional[T] unwraps to T for serializer lookup, so a legal
        # None arrives here with declared=T — never warn on None
        return True
    try:
        return isinstance(value, declared)
    except TypeError:
        return True
