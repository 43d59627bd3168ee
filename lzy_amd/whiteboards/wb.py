"""Whiteboards: versioned result artifacts.

Reference capability (pylzy/lzy/api/v1/whiteboards.py:32-200 +
whiteboards/wrapper.py): ``@whiteboard_(name)`` on a dataclass declares a
whiteboard type; ``wf.create_whiteboard(T, tags=…)`` yields a writable
instance whose fields accept op outputs (lazy proxies) or direct values;
at workflow exit field data is copied to
``whiteboards/{name}/{id}/{field}`` (layout parity with the reference,
api/v1/whiteboards.py:90-92) and the board is FINALIZED in the index.
Reading returns a wrapper that lazily loads fields from storage.

Every field's type must have a *stable* serializer (reference rule,
whiteboards.py:100-110).
"""
from __future__ import annotations

import dataclasses
import datetime
import re
import uuid
from typing import TYPE_CHECKING, Any, Dict, Sequence, Set, Type

from lzy_amd.proxy import get_proxy_entry_id, is_lzy_proxy
from lzy_amd.whiteboards.index import (
    WB_STATUS_CREATED,
    WhiteboardField,
    WhiteboardIndexClient,
    WhiteboardMeta,
)

if TYPE_CHECKING:
    from lzy_amd.core.workflow import LzyWorkflow

WB_NAME_FIELD = "__lzy_wb_name__"
_NAME_RE = re.compile(r"^[A-Za-z0-9_\-]+$")


def whiteboard_(name: str):
    """Class decorator declaring a whiteboard (combine with @dataclass)."""
    if not name or not isinstance(name, str):
        raise TypeError("Whiteboard name must be a non-empty string")
    if not _NAME_RE.match(name):
        raise ValueError("Whiteboard name may contain only [A-Za-z0-9_-]")

    def deco(cls: Type) -> Type:
        setattr(cls, WB_NAME_FIELD, name)
        return cls

    return deco


# alias matching the reference's public name
whiteboard = whiteboard_


def is_whiteboard(typ: Type) -> bool:
    return hasattr(typ, WB_NAME_FIELD) and dataclasses.is_dataclass(typ)


def whiteboard_name(typ: Type) -> str:
    return getattr(typ, WB_NAME_FIELD)


class WritableWhiteboard:
    _internal = {
        "_wb_workflow", "_wb_meta", "_wb_fields", "_wb_assigned", "_wb_typ",
        "_wb_field_entries", "_wb_finalized",
    }

    def __init__(self, typ: Type, tags: Sequence[str], workflow: "LzyWorkflow") -> None:
        if not is_whiteboard(typ):
            raise TypeError(
                "Whiteboard class must be annotated with @whiteboard_ and @dataclass"
            )
        name = whiteboard_name(typ)
        wb_id = str(uuid.uuid4())
        owner = workflow.owner
        storage_uri = f"{owner.storage_uri}/whiteboards/{name}/{wb_id}"

        fields: Dict[str, dataclasses.Field] = {}
        field_entries: Dict[str, str] = {}
        registry = owner.serializer_registry
        # resolve string annotations (PEP 563 `from __future__ import
        # annotations` makes f.type a string)
        try:
            import typing

            hints = typing.get_type_hints(typ)
        except Exception:
            hints = {}
        for f in dataclasses.fields(typ):
            declared = hints.get(f.name, f.type)
            ftyp = declared if isinstance(declared, type) else object
            ser = registry.find_serializer_by_type(ftyp)
            if ser is None:
                raise TypeError(f"Cannot find serializer for whiteboard field {f.name}")
            if not ser.available():
                raise TypeError(
                    f"Serializer for field {f.name} unavailable; install "
                    f"{ser.requirements()}"
                )
            if not ser.stable():
                raise TypeError(
                    f"Field {f.name} of type {f.type} has no stable serializer: "
                    f"whiteboard fields must be portable"
                )
            fields[f.name] = f
            if f.default is not dataclasses.MISSING:
                entry = workflow.snapshot.create_entry(
                    f"wb.{name}.{f.name}.default", ftyp
                )
                workflow.snapshot.put(entry.id, f.default)
                field_entries[f.name] = entry.id

        meta = WhiteboardMeta(
            id=wb_id,
            name=name,
            tags=list(tags),
            status=WB_STATUS_CREATED,
            storage_uri=storage_uri,
            storage_name=owner.storage_name,
            created_at=datetime.datetime.now(datetime.timezone.utc),
            finalized_at=None,
        )
        owner.whiteboard_index.register(meta)

        object.__setattr__(self, "_wb_workflow", workflow)
        object.__setattr__(self, "_wb_meta", meta)
        object.__setattr__(self, "_wb_fields", fields)
        object.__setattr__(self, "_wb_field_entries", field_entries)
        object.__setattr__(self, "_wb_assigned", set(field_entries.keys()))
        object.__setattr__(self, "_wb_typ", typ)
        object.__setattr__(self, "_wb_finalized", False)

    @property
    def id(self) -> str:
        return self._wb_meta.id

    @property
    def name(self) -> str:
        return self._wb_meta.name

    @property
    def tags(self) -> Sequence[str]:
        return tuple(self._wb_meta.tags)

    @property
    def storage_uri(self) -> str:
        return self._wb_meta.storage_uri

    def __setattr__(self, key: str, value: Any) -> None:
        if key in WritableWhiteboard._internal:
            object.__setattr__(self, key, value)
            return
        fields: Dict[str, dataclasses.Field] = self._wb_fields
        if key not in fields:
            raise AttributeError(f"No field {key} on whiteboard {self.name}")
        workflow: "LzyWorkflow" = self._wb_workflow
        if is_lzy_proxy(value):
            entry_id = get_proxy_entry_id(value)
        else:
            ftyp = fields[key].type if isinstance(fields[key].type, type) else object
            entry = workflow.snapshot.create_entry(
                f"wb.{self.name}.{key}", ftyp
            )
            workflow.snapshot.put(entry.id, value)
            entry_id = entry.id
        self._wb_field_entries[key] = entry_id
        self._wb_assigned.add(key)

    def __getattr__(self, key: str) -> Any:
        entries: Dict[str, str] = object.__getattribute__(self, "_wb_field_entries")
        if key in entries:
            wf: "LzyWorkflow" = object.__getattribute__(self, "_wb_workflow")
            got = wf.snapshot.try_get(entries[key])
            if got.found:
                return got.value
            raise AttributeError(f"Field {key} is not materialized yet")
        raise AttributeError(key)

    def _finalize(self) -> None:
        if self._wb_finalized:
            return
        workflow: "LzyWorkflow" = self._wb_workflow
        owner = workflow.owner
        missing = set(self._wb_fields.keys()) - self._wb_assigned
        if missing:
            raise RuntimeError(
                f"Whiteboard {self.name}: fields never assigned: {sorted(missing)}"
            )
        updates: Dict[str, WhiteboardField] = {}
        for fname, entry_id in self._wb_field_entries.items():
            uri = f"{self.storage_uri}/{fname}"
            workflow.snapshot.copy_data(entry_id, uri)
            entry = workflow.snapshot.get_entry(entry_id)
            updates[fname] = WhiteboardField(
                name=fname,
                data_format=entry.data_format,
                type_name=f"{getattr(entry.typ, '__module__', '')}."
                          f"{getattr(entry.typ, '__qualname__', entry.typ)}",
                uri=uri,
            )
        owner.whiteboard_index.update_fields(self.id, updates)
        owner.whiteboard_index.finalize(self.id)
        object.__setattr__(self, "_wb_finalized", True)


class ReadOnlyWhiteboard:
    """Lazy read wrapper (reference whiteboards/wrapper.py)."""

    def __init__(self, meta: WhiteboardMeta, owner) -> None:
        self._meta = meta
        self._owner = owner
        self._cache: Dict[str, Any] = {}

    @property
    def id(self) -> str:
        return self._meta.id

    @property
    def name(self) -> str:
        return self._meta.name

    @property
    def tags(self) -> Sequence[str]:
        return tuple(self._meta.tags)

    @property
    def status(self) -> str:
        return self._meta.status

    @property
    def created_at(self) -> datetime.datetime:
        return self._meta.created_at

    def __getattr__(self, key: str) -> Any:
        meta: WhiteboardMeta = object.__getattribute__(self, "_meta")
        cache = object.__getattribute__(self, "_cache")
        if key in cache:
            return cache[key]
        if key not in meta.fields:
            raise AttributeError(f"No field {key} on whiteboard {meta.name}")
        f = meta.fields[key]
        owner = object.__getattribute__(self, "_owner")
        data = owner.storage_client.read_bytes(f.uri)
        value = owner.serializer_registry.loads(data, f.data_format)
        cache[key] = value
        return value

    def __repr__(self) -> str:
        return f"Whiteboard({self.name}, id={self.id[:8]}, status={self.status})"
