"""Layered configuration (reference: Micronaut YAML + @ConfigurationProperties
per service, e.g. lzy-service/config/LzyServiceConfig.java,
allocator/configs/ServiceConfig.java; test overrides as CLI -key=value args
in test-context/config/LzyConfig.java; Python side env vars in
pylzy/lzy/api/v1/remote/lzy_service_client.py).

Single-node equivalent: one typed config object for the whole runtime,
resolved in precedence order

    explicit overrides  >  environment (LZY_*)  >  YAML file  >  defaults

The YAML file is looked up at ``$LZY_CONFIG`` then ``./lzy.yaml``.  Every
field maps to an env var ``LZY_<FIELD>`` (upper-case).  ``Config.get()``
returns the process-wide instance; tests may ``Config.reset(**overrides)``.
"""
from __future__ import annotations

import os
import threading
from dataclasses import dataclass, field, fields
from typing import Any, Dict, Optional


def _coerce(value: str, typ: type) -> Any:
    if typ is bool:
        return value.strip().lower() in ("1", "true", "yes", "on")
    if typ is int:
        return int(value)
    if typ is float:
        return float(value)
    return value


@dataclass
class Config:
    """All runtime knobs in one place.

    Fields mirror the reference's per-service configs that still make
    sense on one node; the cloud-only ones (K8s credentials, YC disks,
    Kafka brokers, S3 endpoints) have no equivalent here by design.
    """

    # storage (reference: storage vending, lzy-service GetOrCreateDefaultStorage)
    storage: str = ""                  # root URI/path of the durable tier; "" -> tmp default
    # channels (reference: channel-manager + slots transports)
    channel_transport: str = "rccl"    # "rccl" | "ipc"
    channel_chunk_mb: int = 256        # chunk size for large-tensor transfers
    channel_wire_cast: str = ""        # ""|fp16|bf16|fp8e4m3|fp8e5m2: lossy wire dtype
    stream_merge: bool = True          # fold pair_reduce trees into streamed plans
    stream_chunk_mb: int = 32          # chunk size for streamed tree plans (1 GiB shard -> 32 chunks; pipeline fill ~2 chunks per extra level)
    # HIP data-plane kernels
    hip_max_blocks: int = 0            # 0 -> kernel default grid cap
    op_streams: int = 4                # HIP streams per device for op overlap
    # scheduler / pool
    dispatch_workers: int = 0          # 0 -> auto (LocalRuntime thread pool size)
    exec_threads: int = 2              # executor threads per pool rank
    chain_dispatch: bool = True        # eager same-rank dependency dispatch
    gang_timeout_s: float = 120.0      # gang-allocation wait bound
    heartbeat_period_s: float = 2.0    # worker liveness probe period
    task_retries: int = 1              # re-dispatches per task after worker death
    settle_wait_s: float = 120.0       # worker wait for an inbound entry to land
    ack_wait_s: float = 300.0          # driver wait for worker acks
    # result cache / snapshot
    cache_enabled: bool = True
    journal_ttl_hours: float = 168.0   # runtime prunes older journals at start (0 = off)
    keep_hot_max: int = 10000          # post-exit leaf results kept hot (oldest dropped)
    # HBM store spill tier (pinned-host async)
    spill_enabled: bool = True
    spill_threshold_frac: float = 0.85
    # logs / metrics
    log_archive: bool = True           # archive per-op std logs to durable tier
    log_stream_period_s: float = 0.25  # remote-rank live log flush period
    metrics_port: int = 0              # >0 -> serve /metrics on this port
    status_port: int = 0               # >0 -> serve status endpoint on this port
    # client-visible identity (reference: lzy_auth)
    user: str = ""

    _frozen_env: Dict[str, str] = field(default_factory=dict, repr=False)

    # -- resolution ------------------------------------------------------
    @staticmethod
    def _load_yaml() -> Dict[str, Any]:
        path = os.environ.get("LZY_CONFIG") or (
            "lzy.yaml" if os.path.exists("lzy.yaml") else ""
        )
        if not path or not os.path.exists(path):
            return {}
        import yaml

        with open(path) as f:
            data = yaml.safe_load(f) or {}
        if not isinstance(data, dict):
            raise ValueError(f"config file {path} must hold a mapping")
        return data

    @classmethod
    def resolve(cls, **overrides: Any) -> "Config":
        data: Dict[str, Any] = {}
        yaml_data = cls._load_yaml()
        for f in fields(cls):
            if f.name.startswith("_"):
                continue
            if f.name in yaml_data:
                data[f.name] = yaml_data[f.name]
            env_key = f"LZY_{f.name.upper()}"
            if env_key in os.environ:
                data[f.name] = _coerce(os.environ[env_key], f.type if isinstance(f.type, type) else type(f.default))  # type: ignore[arg-type]
            if f.name in overrides and overrides[f.name] is not None:
                data[f.name] = overrides[f.name]
        # back-compat aliases kept from earlier revisions
        if "storage" not in data and os.environ.get("LZY_AMD_STORAGE"):
            data["storage"] = os.environ["LZY_AMD_STORAGE"]
        if "channel_transport" not in data and os.environ.get("LZY_CHANNEL_TRANSPORT"):
            data["channel_transport"] = os.environ["LZY_CHANNEL_TRANSPORT"]
        if "hip_max_blocks" not in data and os.environ.get("LZY_HIP_MAX_BLOCKS"):
            data["hip_max_blocks"] = int(os.environ["LZY_HIP_MAX_BLOCKS"])
        cfg = cls(**data)
        for f in fields(cls):
            if not f.name.startswith("_"):
                v = getattr(cfg, f.name)
                want = f.type if isinstance(f.type, type) else type(f.default)
                if want in (int, float, bool, str) and not isinstance(v, want):
                    setattr(cfg, f.name, _coerce(str(v), want))
        return cfg

    # -- process-wide instance ------------------------------------------
    # The cache is keyed by a fingerprint of the LZY_* environment: a
    # changed env (tests monkeypatching per test; lzy_auth) re-resolves,
    # a stable env costs one dict scan per get().
    _instance: "Optional[Config]" = None
    _fingerprint: Optional[tuple] = None
    _pinned: bool = False
    _lock = threading.Lock()

    @staticmethod
    def _env_fingerprint() -> tuple:
        return tuple(sorted(
            (k, v) for k, v in os.environ.items() if k.startswith("LZY_")
        ))

    @classmethod
    def get(cls) -> "Config":
        with cls._lock:
            fp = cls._env_fingerprint()
            if cls._instance is None or (not cls._pinned and fp != cls._fingerprint):
                cls._instance = cls.resolve()
                cls._fingerprint = fp
                cls._pinned = False
            return cls._instance

    @classmethod
    def reset(cls, **overrides: Any) -> "Config":
        """Re-resolve now; with overrides the result is pinned until the
        next reset (explicit overrides outrank later env edits)."""
        with cls._lock:
            cls._instance = cls.resolve(**overrides)
            cls._fingerprint = cls._env_fingerprint()
            cls._pinned = bool(overrides)
            return cls._instance


def get_config() -> Config:
    return Config.get()
