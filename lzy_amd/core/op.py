"""@op decorator (reference: pylzy/lzy/core/op.py:18-61).

Return-type annotation is mandatory unless ``output_types`` is given —
same contract as the reference.
"""
from __future__ import annotations

from typing import Any, Callable, Optional, Sequence, Type

from lzy_amd.core.call import LazyCallWrapper
from lzy_amd.core.signatures import infer_return_types
from lzy_amd.env.environment import LzyEnvironment
from lzy_amd.env.provisioning import Provisioning


def op(
    func: Optional[Callable] = None,
    *,
    env: Optional[LzyEnvironment] = None,
    output_types: Optional[Sequence[Type]] = None,
    description: str = "",
    version: str = "0.0",
    cache: bool = False,
    lazy_arguments: bool = False,
    gpu_count: Optional[int] = None,
    gpu_type: Optional[str] = None,
    cpu_count: Optional[int] = None,
    ram_size_gb: Optional[int] = None,
    pair_reduce=None,
):
    """Make a function lazily executable inside a workflow.

    ``gpu_count``/``gpu_type``/``cpu_count``/``ram_size_gb`` are provisioning
    shortcuts (reference routes them through env.provisioning; we accept
    them directly too because GPU placement is the common case here).

    ``pair_reduce=(alpha, beta)`` declares the op computes the elementwise
    ``alpha*a + beta*b`` of its two same-shape tensor arguments.  The pool
    runtime may then fold trees of such ops into chunk-streamed reduction
    plans over xGMI (channels/treeplan.py) — the function body remains the
    semantic reference and still runs on the LocalRuntime / same-rank /
    fallback paths.
    """

    def deco(f: Callable) -> LazyCallWrapper:
        out_types = output_types
        if out_types is None:
            inferred = infer_return_types(f)
            if inferred is None:
                raise TypeError(
                    f"Return type is not annotated for {getattr(f, '__qualname__', f)}. "
                    f"Annotate the return type for proper use of @op."
                )
            out_types = inferred

        op_env = env or LzyEnvironment()
        if any(v is not None for v in (gpu_count, gpu_type, cpu_count, ram_size_gb)):
            op_env = op_env.with_provisioning(
                op_env.provisioning.combine(
                    Provisioning(
                        cpu_count=cpu_count,
                        ram_size_gb=ram_size_gb,
                        gpu_count=gpu_count,
                        gpu_type=gpu_type,
                    )
                )
            )

        return LazyCallWrapper(
            function=f,
            output_types=out_types,
            env=op_env,
            description=description,
            version=version,
            cache=cache,
            lazy_arguments=lazy_arguments,
            pair_reduce=tuple(pair_reduce) if pair_reduce is not None else None,
        )

    if func is None:
        return deco
    return deco(func)
