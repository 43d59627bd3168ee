"""Materialize-on-touch lazy proxies.

Same capability as the reference's Proxifier metaclass machinery
(/root/reference/pylzy/lzy/proxy/automagic.py:29-107 and
pylzy/lzy/api/v1/utils/proxy_adapter.py:16-83), re-designed: instead of a
metaclass that re-creates the target type, we synthesize one forwarding class
per proxy whose dunder methods materialize the underlying value on first
touch and delegate to it.  The materialized value is cached on the class
(one class per proxy instance), so repeated touches are a dict hit.

Special cases kept from the reference semantics:
  * ``None``-able results: an op annotated ``-> Optional[T]`` may return
    None; ``__lzy_origin__`` then IS None and boolean tests work
    (reference: pylzy/lzy/core/call.py:234-250).
  * proxies survive pickling by reducing to the materialized value
    (reference registers a copyreg reducer; we implement ``__reduce_ex__``).
"""
from __future__ import annotations

from typing import Any, Callable, Sequence, Tuple, Type

_PROXIED_FLAG = "__lzy_proxied__"
_ENTRY_ID_ATTR = "__lzy_entry_id__"

# Dunders that must be defined on the *class* to take part in the C-level
# protocol lookups (Python skips instance __getattr__ for these).
_FORWARDED_DUNDERS = [
    "__abs__", "__add__", "__and__", "__bool__", "__call__", "__ceil__",
    "__contains__", "__delitem__", "__divmod__", "__enter__", "__eq__",
    "__exit__", "__float__", "__floor__", "__floordiv__", "__ge__",
    "__getitem__", "__gt__", "__hash__", "__iadd__", "__iand__",
    "__ifloordiv__", "__ilshift__", "__imod__", "__imul__", "__index__",
    "__int__", "__invert__", "__ior__", "__ipow__", "__irshift__",
    "__isub__", "__iter__", "__itruediv__", "__ixor__", "__le__",
    "__len__", "__lshift__", "__lt__", "__matmul__", "__mod__", "__mul__",
    "__ne__", "__neg__", "__next__", "__or__", "__pos__", "__pow__",
    "__radd__", "__rand__", "__rdivmod__", "__rfloordiv__", "__rlshift__",
    "__rmatmul__", "__rmod__", "__rmul__", "__ror__", "__round__",
    "__rpow__", "__rrshift__", "__rshift__", "__rsub__", "__rtruediv__",
    "__rxor__", "__setitem__", "__str__", "__repr__", "__sub__",
    "__truediv__", "__trunc__", "__xor__", "__format__", "__length_hint__",
    "__reversed__", "__bytes__",
]


def _make_forwarder(name: str) -> Callable:
    def forward(self, *args, **kwargs):
        origin = type(self).__lzy_materialize__()
        return getattr(origin, name)(*args, **kwargs)

    forward.__name__ = name
    return forward


def proxy(
    materializer: Callable[[], Any],
    types: Sequence[Type],
    cls_attrs: dict | None = None,
) -> Any:
    """Build a lazy proxy claiming to be of ``types``.

    ``materializer`` is called once, on first touch; its result is cached.
    """
    cls_attrs = dict(cls_attrs or {})
    state: dict = {"done": False, "value": None}

    def __lzy_materialize__() -> Any:
        if not state["done"]:
            state["value"] = materializer()
            state["done"] = True
        return state["value"]

    ns: dict = {name: _make_forwarder(name) for name in _FORWARDED_DUNDERS}

    def __getattr__(self, item):  # noqa: N807 - proxy forwarding
        origin = type(self).__lzy_materialize__()
        return getattr(origin, item)

    def __setattr__(self, item, value):
        origin = type(self).__lzy_materialize__()
        setattr(origin, item, value)

    def __instancecheck_getter(self):
        return type(type(self).__lzy_materialize__())

    def __reduce_ex__(self, protocol):
        # Pickling a proxy pickles the materialized value (the reference
        # installs a copyreg reducer for the same effect).
        origin = type(self).__lzy_materialize__()
        import cloudpickle

        return (cloudpickle.loads, (cloudpickle.dumps(origin),))

    ns.update(
        {
            "__getattr__": __getattr__,
            "__setattr__": __setattr__,
            "__reduce_ex__": __reduce_ex__,
            "__lzy_materialize__": staticmethod(__lzy_materialize__),
            "__lzy_proxy_types__": tuple(types),
            "__class_getitem__": classmethod(lambda cls, item: cls),
            _PROXIED_FLAG: True,
        }
    )
    ns.update(cls_attrs)

    # Properties the adapter API exposes (match reference names:
    # proxy_adapter.py:28-33).
    ns["__lzy_materialized__"] = property(lambda self: state["done"])
    ns["__lzy_origin__"] = property(lambda self: __lzy_materialize__())
    # isinstance(proxy, T) reflects the MATERIALIZED value's type —
    # CPython's isinstance consults __class__ when the C-level type
    # check fails (reference: test_simple_isinstance/.._none semantics:
    # a None-valued Optional proxy is an instance of NoneType, not T).
    ns["__class__"] = property(__instancecheck_getter)

    type_names = "_".join(getattr(t, "__name__", "obj") for t in types) or "obj"
    cls = type(f"LzyProxy_{type_names}", (), ns)
    return cls()


# ---------------------------------------------------------------------------
# Adapter API (reference: pylzy/lzy/api/v1/utils/proxy_adapter.py)
# ---------------------------------------------------------------------------

def is_lzy_proxy(obj: Any) -> bool:
    cls = type(obj)
    return bool(getattr(cls, _PROXIED_FLAG, False))


def get_proxy_entry_id(obj: Any) -> str:
    if not is_lzy_proxy(obj):
        raise ValueError(f"Object {obj!r} is not a lazy proxy")
    return getattr(type(obj), _ENTRY_ID_ATTR)


def materialized(obj: Any) -> bool:
    return bool(obj.__lzy_materialized__)


def materialize(obj: Any) -> Any:
    return obj.__lzy_origin__


def materialize_if_sequence_of_proxies(obj: Any) -> Any:
    if not isinstance(obj, (tuple, list)) or len(obj) == 0 or not is_lzy_proxy(obj[0]):
        return obj
    out = [materialize(e) if is_lzy_proxy(e) else e for e in obj]
    return tuple(out) if isinstance(obj, tuple) else out


def lzy_proxy(entry_id: str, types: Sequence[Type], workflow: Any, known_value: Any = None,
              has_value: bool = False) -> Any:
    """Proxy bound to a workflow snapshot entry.

    On touch: return the known value, else read the entry from the store,
    else run the workflow barrier and read again
    (reference: proxy_adapter.py:56-83).
    """

    def _materialize() -> Any:
        if has_value:
            return known_value
        got = workflow.snapshot.try_get(entry_id)
        if got.found:
            return got.value
        workflow.barrier()
        got = workflow.snapshot.try_get(entry_id)
        if got.found:
            return got.value
        raise RuntimeError(
            f"Cannot materialize entry {entry_id} of workflow {workflow.name}"
        )

    return proxy(_materialize, types, cls_attrs={_ENTRY_ID_ATTR: entry_id})
