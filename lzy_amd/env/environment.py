"""Composable execution environment.

Reference capability (pylzy/lzy/env/environment.py:27 + env/mixin.py):
an env object carries provisioning + python-env + container + env-vars and
merges Lzy -> workflow -> op.  MI355X re-design: workers are in-process on
the same node, so there is no conda/docker sync — the python env IS the
driver's env (that's the whole point: zero provisioning latency vs the
reference's conda create/update per VM).  We keep env_variables (applied
around op execution) and provisioning (GPU placement).  Container/python
fields are accepted for source compatibility and validated as no-ops.
"""
from __future__ import annotations

from dataclasses import dataclass, field, replace
from typing import Any, Dict, Mapping, Optional

from lzy_amd.env.provisioning import Provisioning


@dataclass(frozen=True)
class LzyEnvironment:
    env_variables: Dict[str, str] = field(default_factory=dict)
    provisioning: Provisioning = field(default_factory=Provisioning)
    namespace: Dict[str, Any] = field(default_factory=dict)

    def with_fields(self, **kwargs: Any) -> "LzyEnvironment":
        return replace(self, **kwargs)

    def with_env_variables(self, **env_vars: str) -> "LzyEnvironment":
        return self.with_fields(env_variables={**self.env_variables, **env_vars})

    def with_provisioning(self, provisioning: Provisioning) -> "LzyEnvironment":
        return self.with_fields(provisioning=provisioning)

    def get_namespace(self) -> Dict[str, Any]:
        return self.namespace

    def combine(self, other: "LzyEnvironment") -> "LzyEnvironment":
        """other overrides self field-wise (op > workflow > Lzy)."""
        return LzyEnvironment(
            env_variables={**self.env_variables, **other.env_variables},
            provisioning=self.provisioning.combine(other.provisioning),
            namespace={**self.namespace, **other.namespace},
        )

    def __call__(self, subject):
        """Reference applier spelling (pylzy/lzy/env/shortcuts.py:28-103
        EnvironmentApplierType): shortcuts stack as decorators ABOVE
        @op — ``@provisioning(cpu_count=8)`` / ``@gpu(2)`` over an op
        wrapper merges this env into it (this env's fields win)."""
        if hasattr(subject, "with_env") and hasattr(subject, "env"):
            return subject.with_env(subject.env.combine(self))
        raise TypeError(
            "an LzyEnvironment shortcut used as a decorator must wrap an "
            "@op wrapper (or another env-bearing object) — put it ABOVE "
            "@op, e.g.\n  @provisioning(cpu_count=8)\n  @op\n  def f(): ..."
        )


class WithEnvironmentMixin:
    """Fluent env shortcuts shared by Lzy / workflow / op wrappers
    (reference: pylzy/lzy/env/mixin.py)."""

    env: LzyEnvironment

    def with_fields(self, **kwargs: Any):
        from dataclasses import replace as _replace

        return _replace(self, **kwargs)

    def with_env(self, env: LzyEnvironment):
        return self.with_fields(env=env)

    def with_env_variables(self, **env_vars: str):
        return self.with_env(self.env.with_env_variables(**env_vars))

    def with_provisioning(
        self,
        *,
        cpu_count: Optional[int] = None,
        ram_size_gb: Optional[int] = None,
        gpu_count: Optional[int] = None,
        gpu_type: Optional[str] = None,
    ):
        return self.with_env(
            self.env.with_provisioning(
                self.env.provisioning.combine(
                    Provisioning(cpu_count, ram_size_gb, gpu_count, gpu_type)
                )
            )
        )
