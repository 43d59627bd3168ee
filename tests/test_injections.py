"""CatBoost injection tests against a stub catboost module (the real
library is optional — the reference's integration is equally lazy:
pylzy/lzy/injections/catboost.py:13).  The stub exercises the actual
injection mechanics: fit patched, provisioned fit routed through an @op,
trained state adopted back."""
import sys
import types

import pytest


@pytest.fixture()
def stub_catboost(monkeypatch):
    mod = types.ModuleType("catboost")

    class CatBoost:
        def __init__(self):
            self.fitted_with = None
            self.is_fitted = False

        def fit(self, X, y=None, **kwargs):
            self.fitted_with = (X, y, kwargs)
            self.is_fitted = True
            return self

    mod.CatBoost = CatBoost
    monkeypatch.setitem(sys.modules, "catboost", mod)
    yield mod


def test_inject_patches_once(stub_catboost):
    from lzy_amd.injections.catboost import inject_catboost

    inject_catboost()
    assert stub_catboost.CatBoost.__lzy_injected__
    patched = stub_catboost.CatBoost.fit
    inject_catboost()  # idempotent
    assert stub_catboost.CatBoost.fit is patched


def test_unprovisioned_fit_runs_locally(stub_catboost, lzy):
    from lzy_amd.injections.catboost import inject_catboost

    inject_catboost()
    m = stub_catboost.CatBoost()
    m.fit([[1, 2]], [0])
    assert m.is_fitted
    assert m.fitted_with[2] == {}


def test_provisioned_fit_routes_through_op(stub_catboost, lzy):
    from lzy_amd.env.provisioning import Provisioning
    from lzy_amd.injections.catboost import inject_catboost
    from lzy_amd.utils.metrics import METRICS

    inject_catboost()
    m = stub_catboost.CatBoost()
    # CPU provisioning on this box (GPU provisioning is covered by the
    # gpu-marked pool tests); still must route through an @op
    m.__lzy_provisioning__ = Provisioning(cpu_count=2)
    def fit_op_runs():
        return sum(
            v for (name, labels), v in METRICS._counters.items()
            if name == "lzy_op_runs" and any("fit_op" in lv for _, lv in labels)
        )

    before = fit_op_runs()
    with lzy.workflow("cb", interactive=False):
        m.fit([[1, 2], [3, 4]], [0, 1])
    assert m.is_fitted
    assert fit_op_runs() == before + 1
