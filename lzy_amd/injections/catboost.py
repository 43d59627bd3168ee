"""CatBoost integration.

Reference capability (pylzy/lzy/injections/catboost.py:13-80): monkey-patch
``CatBoost.fit`` so that, when the model was given a provisioning spec,
the fit call runs as a remote @op on the requested hardware.  Here the
"remote hardware" is a GPU rank of the pool (or the local runtime's GPU):
``inject_catboost()`` patches fit to execute as an op inside the active
workflow, with ``task_type="GPU"`` when a GPU was provisioned.

catboost is an optional dependency: importing this module without it
raises ImportError, matching the reference's lazy injection behavior.

VERIFICATION STATUS (honest): the catboost wheel is NOT present in this
image's offline wheelhouse and the build/GPU environments have no
network, so this injection has never been exercised against the real
library — only against the structural stub in tests/test_injections.py
(which drives the actual patching logic: provisioning detection,
task_type="GPU" selection, op capture inside a workflow).  The
BASELINE.json config-2 benchmark therefore uses a torch GBDT-style
stand-in objective, explicitly labeled as such in
benchmarks/baseline_configs.py.  First action in an environment where
``pip install catboost`` works: run
``python -m pytest tests/test_injections.py`` with the real module on
the path (the stub fixture yields to a real import automatically) and
benchmarks/baseline_configs.py --config 2.
"""
from __future__ import annotations

from typing import Any, Optional

from lzy_amd.core.op import op
from lzy_amd.env.provisioning import Provisioning


def inject_catboost() -> None:
    from catboost import CatBoost  # noqa: F401 - optional dep

    if getattr(CatBoost, "__lzy_injected__", False):
        return

    original_fit = CatBoost.fit

    def provisioned_fit(self, *args: Any, **kwargs: Any):
        prov: Optional[Provisioning] = getattr(self, "__lzy_provisioning__", None)
        from lzy_amd.core.workflow import LzyWorkflow

        wf = LzyWorkflow.get_active()
        if prov is None or wf is None:
            return original_fit(self, *args, **kwargs)

        gpus = prov.effective_gpu_count

        def fit_op(model: Any, fit_args: tuple, fit_kwargs: dict) -> Any:
            if gpus > 0:
                fit_kwargs.setdefault("task_type", "GPU")
                fit_kwargs.setdefault("devices", "0")
            original_fit(model, *fit_args, **fit_kwargs)
            return model

        fitted = op(
            fit_op,
            output_types=(type(self),),
            gpu_count=gpus if gpus > 0 else None,
        )(self, args, kwargs)
        # materialize and adopt the trained state (reference returns the
        # fitted model from the remote op the same way)
        from lzy_amd.proxy import materialize, is_lzy_proxy

        trained = materialize(fitted) if is_lzy_proxy(fitted) else fitted
        self.__dict__.update(trained.__dict__)
        return self

    def on_gpu(self, gpu_count: int = 1, gpu_type: str = "MI355X"):
        self.__lzy_provisioning__ = Provisioning(gpu_count=gpu_count, gpu_type=gpu_type)
        return self

    CatBoost.fit = provisioned_fit
    CatBoost.on_gpu = on_gpu
    CatBoost.__lzy_injected__ = True
