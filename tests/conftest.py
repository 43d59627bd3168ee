import os
import tempfile

import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")


@pytest.fixture()
def storage_root(tmp_path, monkeypatch):
    """Isolated storage root per test."""
    from lzy_amd.runtime.local import DEFAULT_STORAGE_ENV

    monkeypatch.setenv(DEFAULT_STORAGE_ENV, str(tmp_path / "storage"))
    return tmp_path / "storage"


@pytest.fixture()
def lzy(storage_root):
    from lzy_amd import Lzy
    from lzy_amd.runtime.local import LocalRuntime

    return Lzy(runtime=LocalRuntime())

