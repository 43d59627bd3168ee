"""Lazy call capture.

Reference capability (pylzy/lzy/core/call.py:40-268): calling a decorated
function inside a workflow builds an ``LzyCall`` — snapshot entries are
created for every argument, return value and the (possible) exception;
argument *values* are registered immediately (the reference uploads them
to S3 eagerly; we register them in the in-process store — zero copies);
the call returns lazy proxies for its outputs.
"""
from __future__ import annotations

import logging
import uuid
from dataclasses import dataclass
from typing import TYPE_CHECKING, Any, Dict, Optional, Sequence, Tuple, Type

from lzy_amd.core.signatures import (
    CallSignature,
    check_type_compatible,
    infer_call_signature,
)
from lzy_amd.env.environment import LzyEnvironment, WithEnvironmentMixin
from lzy_amd.proxy import get_proxy_entry_id, is_lzy_proxy, lzy_proxy

if TYPE_CHECKING:
    from lzy_amd.core.workflow import LzyWorkflow

_LOG = logging.getLogger("lzy_amd.call")


class LzyCall:
    def __init__(
        self,
        workflow: "LzyWorkflow",
        signature: CallSignature,
        output_types: Sequence[Type],
        env: LzyEnvironment,
        description: str = "",
        version: str = "0.0",
        cache: bool = False,
        lazy_arguments: bool = False,
        pair_reduce=None,
    ) -> None:
        from lzy_amd.utils.ids import fast_uid

        self.id = fast_uid()
        self.workflow = workflow
        self.signature = signature
        self.output_types = tuple(output_types)
        self.env = env
        self.description = description
        self.version = version
        self.cache = cache
        self.lazy_arguments = lazy_arguments
        # (alpha, beta) for ops declared as pairwise tensor reductions —
        # lets the pool runtime fold merge trees into streamed plans
        self.pair_reduce = pair_reduce

        snapshot = workflow.snapshot
        name = signature.qualname

        # argument entries; values registered immediately (reference
        # call.py:62-100 uploads eagerly — here it's a dict insert, and a
        # device tensor arg stays exactly where it is in HBM).
        self.arg_entry_ids: Tuple[str, ...] = tuple(
            self._entry_for_value(v, t, f"{name}.arg[{n}]")
            for v, t, n in zip(signature.args, signature.arg_types, signature.arg_names)
        )
        self.kwarg_entry_ids: Dict[str, str] = {
            k: self._entry_for_value(v, signature.kwarg_types.get(k, object), f"{name}.kwarg[{k}]")
            for k, v in signature.kwargs.items()
        }

        self.entry_ids: Tuple[str, ...] = tuple(
            snapshot.create_entry(f"{name}.return[{i}]", t).id
            for i, t in enumerate(self.output_types)
        )
        self.exception_id: str = snapshot.create_entry(f"{name}.exception", tuple).id

    def _entry_for_value(self, value: Any, declared: Type, name: str) -> str:
        if is_lzy_proxy(value):
            return get_proxy_entry_id(value)
        if not check_type_compatible(value, declared):
            _LOG.warning(
                "Argument %s: declared type %s does not match actual %s",
                name, declared, type(value),
            )
        entry = self.workflow.snapshot.create_entry(name, type(value))
        self.workflow.snapshot.put(entry.id, value)
        return entry.id

    @property
    def callable_name(self) -> str:
        return self.signature.qualname

    def input_entry_ids(self) -> Tuple[str, ...]:
        return self.arg_entry_ids + tuple(self.kwarg_entry_ids.values())

    def __repr__(self) -> str:
        return f"LzyCall({self.callable_name}, id={self.id[:8]})"


class LazyCallWrapper(WithEnvironmentMixin):
    """What @op returns (reference call.py:192-268): callable that captures
    into the active workflow, or runs the function directly outside one."""

    def __init__(
        self,
        function,
        output_types: Sequence[Type],
        env: LzyEnvironment,
        description: str = "",
        version: str = "0.0",
        cache: bool = False,
        lazy_arguments: bool = False,
        pair_reduce=None,
    ) -> None:
        self.function = function
        self.output_types = tuple(output_types)
        self.env = env
        self.description = description
        self.version = version
        self.cache = cache
        self.lazy_arguments = lazy_arguments
        self.pair_reduce = pair_reduce
        # look like the wrapped function
        self.__name__ = getattr(function, "__name__", "op")
        self.__qualname__ = getattr(function, "__qualname__", self.__name__)
        self.__doc__ = getattr(function, "__doc__", None)

    def with_fields(self, **kwargs: Any) -> "LazyCallWrapper":
        merged = dict(
            function=self.function,
            output_types=self.output_types,
            env=self.env,
            description=self.description,
            version=self.version,
            cache=self.cache,
            lazy_arguments=self.lazy_arguments,
            pair_reduce=self.pair_reduce,
        )
        merged.update(kwargs)
        return LazyCallWrapper(**merged)

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        from lzy_amd.core.workflow import LzyWorkflow

        workflow = LzyWorkflow.get_active()
        if workflow is None:
            # outside a workflow the op is just the function (reference parity)
            return self.function(*args, **kwargs)

        signature = infer_call_signature(self.function, *args, **kwargs)
        env = workflow.env.combine(self.env)
        call = LzyCall(
            workflow=workflow,
            signature=signature,
            output_types=self.output_types,
            env=env,
            description=self.description,
            version=self.version,
            cache=self.cache,
            lazy_arguments=self.lazy_arguments,
            pair_reduce=self.pair_reduce,
        )
        workflow.register_call(call)

        proxies = tuple(
            lzy_proxy(eid, (typ,), workflow)
            for eid, typ in zip(call.entry_ids, call.output_types)
        )
        if len(proxies) == 1:
            return proxies[0]
        return proxies
