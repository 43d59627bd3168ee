"""Provisioning requirements and pool resolution.

Reference capability (pylzy/lzy/env/provisioning/provisioning.py:60-126):
an op declares cpu/gpu/ram requirements and a score function matches them
against available VM pools.  MI355X re-design: the "pools" are the GPUs of
one node (8 × MI355X, 288 GB HBM3E each) plus the host CPU; resolution
picks a *placement class* — how many GPUs the op needs gang-scheduled —
and validates it against the runtime's world size.  Scoring degenerates to
exact feasibility checks because the node is homogeneous.
"""
from __future__ import annotations

import enum
from dataclasses import dataclass, field
from typing import Optional, Sequence

from lzy_amd.exceptions import BadProvisioningError


class GpuType(enum.Enum):
    NO_GPU = "NO_GPU"
    MI355X = "MI355X"
    # Reference pool labels kept for source compatibility of user scripts
    # (reference: lzy/allocator .../vmpool/GpuTypes.java:3-8); they all map
    # onto the node's MI355X pool.
    V100 = "V100"
    A100 = "A100"
    T4 = "T4"


class AnyRequirement:
    """'Any value is fine' sentinel (reference: provisioning.py:43
    ``Any = AnyRequirement()``).  Our Provisioning uses None for the
    same meaning; shortcuts normalize this sentinel to None so
    reference scripts that pass ``Any`` run unchanged."""

    _instance: Optional["AnyRequirement"] = None

    def __new__(cls) -> "AnyRequirement":
        if cls._instance is None:
            cls._instance = super().__new__(cls)
        return cls._instance

    def __repr__(self) -> str:
        return "Any"


Any = AnyRequirement()


def _norm(v):
    """Map the Any sentinel (and reference NotSpecified-style Nones) to
    this module's None convention."""
    return None if isinstance(v, AnyRequirement) else v


def maximum_score_function(requested: "Provisioning", pool: "PoolSpec") -> float:
    """Prefer the LARGEST feasible pool (reference: env/provisioning/
    score.py:16 — biggest headroom over the request)."""
    return float(pool.gpu_count * 1024 + pool.cpu_count)


def minimum_score_function(requested: "Provisioning", pool: "PoolSpec") -> float:
    """Prefer the TIGHTEST feasible fit (reference: score.py:28 — least
    over-allocation; the default ranking in resolve_pool)."""
    return -float(pool.gpu_count * 1024 + pool.cpu_count)


@dataclass(frozen=True)
class PoolSpec:
    """One placement class offered by the node."""

    pool_name: str
    cpu_count: int
    ram_size_gb: int
    gpu_count: int
    gpu_type: str


@dataclass(frozen=True)
class Provisioning:
    cpu_count: Optional[int] = None
    ram_size_gb: Optional[int] = None
    gpu_count: Optional[int] = None
    gpu_type: Optional[str] = None

    def combine(self, other: "Provisioning") -> "Provisioning":
        """other's fields override ours (op > workflow > Lzy)."""
        return Provisioning(
            cpu_count=other.cpu_count if other.cpu_count is not None else self.cpu_count,
            ram_size_gb=other.ram_size_gb if other.ram_size_gb is not None else self.ram_size_gb,
            gpu_count=other.gpu_count if other.gpu_count is not None else self.gpu_count,
            gpu_type=other.gpu_type if other.gpu_type is not None else self.gpu_type,
        )

    @property
    def effective_gpu_count(self) -> int:
        if self.gpu_count is not None:
            return self.gpu_count
        if self.gpu_type is not None and self.gpu_type != GpuType.NO_GPU.value:
            return 1
        return 0

    def resolve_pool(self, pools: Sequence[PoolSpec], score=None) -> PoolSpec:
        """Pick the best feasible pool (reference: score-based resolve,
        provisioning.py:126).  ``score(provisioning, pool) -> float`` may
        override the default cheapest-fit ranking (higher wins), matching
        the reference's pluggable ``score`` function."""
        need_gpus = self.effective_gpu_count
        feasible = [
            p
            for p in pools
            if p.gpu_count >= need_gpus
            and (self.cpu_count is None or p.cpu_count >= self.cpu_count)
            and (self.ram_size_gb is None or p.ram_size_gb >= self.ram_size_gb)
        ]
        if not feasible:
            raise BadProvisioningError(
                f"No pool satisfies provisioning {self} (available: {list(pools)})"
            )
        if score is not None:
            return max(feasible, key=lambda p: score(self, p))
        # cheapest = fewest GPUs then fewest CPUs
        return min(feasible, key=lambda p: (p.gpu_count, p.cpu_count))


def node_pools(n_gpus: int, cpu_count: int = 64, ram_gb: int = 1024) -> Sequence[PoolSpec]:
    """The placement classes of this node: CPU-only plus 1..n_gpus MI355X."""
    pools = [PoolSpec("cpu", cpu_count, ram_gb, 0, GpuType.NO_GPU.value)]
    k = 1
    while k <= n_gpus:
        pools.append(PoolSpec(f"mi355x-x{k}", cpu_count, ram_gb, k, GpuType.MI355X.value))
        k *= 2
    return pools
