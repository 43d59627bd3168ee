"""Config layering tests (reference: Micronaut YAML config + CLI overrides,
test-context/config/LzyConfig.java; env-var config lzy_service_client.py)."""
import os

import pytest

from lzy_amd.config import Config


@pytest.fixture(autouse=True)
def _isolate_config(monkeypatch):
    for k in list(os.environ):
        if k.startswith("LZY_"):
            monkeypatch.delenv(k, raising=False)
    yield
    os.environ.pop("LZY_CONFIG", None)
    Config.reset()


def test_defaults():
    cfg = Config.resolve()
    assert cfg.channel_transport == "rccl"
    assert cfg.cache_enabled is True
    assert cfg.hip_max_blocks == 0


def test_env_overrides(monkeypatch):
    monkeypatch.setenv("LZY_CHANNEL_TRANSPORT", "ipc")
    monkeypatch.setenv("LZY_HIP_MAX_BLOCKS", "1024")
    monkeypatch.setenv("LZY_CACHE_ENABLED", "false")
    cfg = Config.resolve()
    assert cfg.channel_transport == "ipc"
    assert cfg.hip_max_blocks == 1024
    assert cfg.cache_enabled is False


def test_yaml_file(tmp_path, monkeypatch):
    p = tmp_path / "lzy.yaml"
    p.write_text("storage: /data/lzy\nchannel_chunk_mb: 64\nuser: alice\n")
    monkeypatch.setenv("LZY_CONFIG", str(p))
    cfg = Config.resolve()
    assert cfg.storage == "/data/lzy"
    assert cfg.channel_chunk_mb == 64
    assert cfg.user == "alice"


def test_precedence_env_over_yaml_explicit_over_env(tmp_path, monkeypatch):
    p = tmp_path / "lzy.yaml"
    p.write_text("channel_chunk_mb: 64\n")
    monkeypatch.setenv("LZY_CONFIG", str(p))
    monkeypatch.setenv("LZY_CHANNEL_CHUNK_MB", "128")
    assert Config.resolve().channel_chunk_mb == 128
    assert Config.resolve(channel_chunk_mb=32).channel_chunk_mb == 32


def test_back_compat_storage_alias(monkeypatch):
    monkeypatch.setenv("LZY_AMD_STORAGE", "/tmp/alias_root")
    assert Config.resolve().storage == "/tmp/alias_root"


def test_process_wide_env_tracking(monkeypatch):
    Config.reset()
    a = Config.get()
    assert Config.get() is a  # stable env -> cached instance
    monkeypatch.setenv("LZY_USER", "bob")
    assert Config.get().user == "bob"  # env change re-resolves


def test_reset_overrides_pin_until_next_reset(monkeypatch):
    Config.reset(channel_chunk_mb=48)
    monkeypatch.setenv("LZY_CHANNEL_CHUNK_MB", "96")
    assert Config.get().channel_chunk_mb == 48  # explicit override pinned
    Config.reset()
    assert Config.get().channel_chunk_mb == 96


def test_bad_yaml_rejected(tmp_path, monkeypatch):
    p = tmp_path / "bad.yaml"
    p.write_text("- just\n- a list\n")
    monkeypatch.setenv("LZY_CONFIG", str(p))
    with pytest.raises(ValueError, match="mapping"):
        Config.resolve()


def test_unknown_yaml_keys_ignored(tmp_path, monkeypatch):
    p = tmp_path / "extra.yaml"
    p.write_text("storage: /x\nnot_a_real_knob: 1\n")
    monkeypatch.setenv("LZY_CONFIG", str(p))
    cfg = Config.resolve()
    assert cfg.storage == "/x"
    assert not hasattr(cfg, "not_a_real_knob")


def test_bool_coercions(monkeypatch):
    for raw, want in [("1", True), ("true", True), ("on", True),
                      ("0", False), ("no", False), ("off", False)]:
        monkeypatch.setenv("LZY_CACHE_ENABLED", raw)
        assert Config.resolve().cache_enabled is want
