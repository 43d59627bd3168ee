"""Storage garbage collection.

Reference capability: GC of stale executions and expired VMs
(lzy-service gc/GarbageCollector.java:21, allocator gc/*): here the
collectible state is the durable tier — per-run snapshots and archived
logs.  Cache blobs and whiteboards are retained by default (they are the
long-lived artifacts); pass flags to collect them too.
"""
from __future__ import annotations

import shutil
import time
from pathlib import Path
from typing import Dict


def gc_storage(
    storage_root: str,
    ttl_seconds: float = 7 * 24 * 3600,
    collect_cache: bool = False,
    collect_whiteboards: bool = False,
    now: float | None = None,
) -> Dict[str, int]:
    """Remove expired run data under ``storage_root``; returns counts."""
    now = now if now is not None else time.time()
    root = Path(storage_root)
    removed = {"runs": 0, "logs": 0, "cache": 0, "whiteboards": 0}

    def _sweep(subdir: str, key: str) -> None:
        base = root / subdir
        if not base.is_dir():
            return
        for child in base.iterdir():
            try:
                if now - child.stat().st_mtime > ttl_seconds:
                    shutil.rmtree(child, ignore_errors=True)
                    removed[key] += 1
            except FileNotFoundError:
                continue

    _sweep("lzy_runs", "runs")
    _sweep("lzy_logs", "logs")
    if collect_cache:
        _sweep("lzy_cache", "cache")
    if collect_whiteboards:
        _sweep("whiteboards", "whiteboards")
    return removed


def gc_journals(
    journal_dir: str,
    ttl_seconds: float = 7 * 24 * 3600,
    now: float | None = None,
) -> int:
    """Remove expired crash-resume journals (one per execution)."""
    import os

    now = now if now is not None else time.time()
    base = Path(journal_dir)
    n = 0
    if not base.is_dir():
        return 0
    for child in base.glob("*.jsonl"):
        try:
            if now - child.stat().st_mtime > ttl_seconds:
                os.unlink(child)
                n += 1
        except FileNotFoundError:
            continue
    return n


def _main() -> None:  # pragma: no cover - thin CLI
    import argparse
    import json
    import os
    import tempfile

    ap = argparse.ArgumentParser(description="lzy_amd storage GC")
    ap.add_argument("--root", default=None, help="storage root (default: config)")
    ap.add_argument("--ttl-hours", type=float, default=7 * 24.0)
    ap.add_argument("--cache", action="store_true", help="also collect result cache")
    ap.add_argument("--whiteboards", action="store_true")
    args = ap.parse_args()
    root = args.root
    if root is None:
        from lzy_amd.config import get_config

        root = get_config().storage or os.path.join(
            tempfile.gettempdir(), "lzy_amd_storage"
        )
        if root.startswith("file://"):
            root = root[len("file://"):]
    removed = gc_storage(
        root, ttl_seconds=args.ttl_hours * 3600,
        collect_cache=args.cache, collect_whiteboards=args.whiteboards,
    )
    removed["journals"] = gc_journals(
        os.path.join(tempfile.gettempdir(), "lzy_amd_journal"),
        ttl_seconds=args.ttl_hours * 3600,
    )
    print(json.dumps(removed))


if __name__ == "__main__":
    _main()
