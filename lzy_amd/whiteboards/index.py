"""Whiteboard metadata index.

Reference capability (lzy/whiteboard WhiteboardService.java:45-261 + pylzy
whiteboards/index.py:47-147): register CREATED, update fields, FINALIZE,
get by id, list by name/tags/time-range.  Single-node re-design: a sqlite
database beside the storage root replaces the Java service + Postgres —
same queries, zero RPC.
"""
from __future__ import annotations

import datetime
import json
import sqlite3
import threading
from dataclasses import dataclass, field
from typing import Any, Dict, Iterable, List, Optional, Sequence

WB_STATUS_CREATED = "CREATED"
WB_STATUS_FINALIZED = "FINALIZED"


@dataclass
class WhiteboardField:
    name: str
    data_format: str
    type_name: str
    uri: str


@dataclass
class WhiteboardMeta:
    id: str
    name: str
    tags: List[str]
    status: str
    storage_uri: str
    storage_name: str
    created_at: datetime.datetime
    finalized_at: Optional[datetime.datetime]
    fields: Dict[str, WhiteboardField] = field(default_factory=dict)


class WhiteboardIndexClient:
    """sqlite-backed index; thread-safe via one connection per call."""

    def __init__(self, db_path: str) -> None:
        self._db_path = db_path
        self._lock = threading.Lock()
        self._init_db()

    def _conn(self) -> sqlite3.Connection:
        conn = sqlite3.connect(self._db_path, timeout=30)
        conn.execute("PRAGMA journal_mode=WAL")
        return conn

    def _init_db(self) -> None:
        with self._lock, self._conn() as conn:
            conn.execute(
                """CREATE TABLE IF NOT EXISTS whiteboards (
                       id TEXT PRIMARY KEY,
                       name TEXT NOT NULL,
                       tags TEXT NOT NULL,
                       status TEXT NOT NULL,
                       storage_uri TEXT NOT NULL,
                       storage_name TEXT NOT NULL,
                       created_at TEXT NOT NULL,
                       finalized_at TEXT,
                       fields TEXT NOT NULL
                   )"""
            )
            conn.execute(
                "CREATE INDEX IF NOT EXISTS wb_name_idx ON whiteboards(name)"
            )

    def register(self, meta: WhiteboardMeta) -> None:
        with self._lock, self._conn() as conn:
            conn.execute(
                "INSERT INTO whiteboards VALUES (?,?,?,?,?,?,?,?,?)",
                (
                    meta.id,
                    meta.name,
                    json.dumps(meta.tags),
                    meta.status,
                    meta.storage_uri,
                    meta.storage_name,
                    meta.created_at.isoformat(),
                    meta.finalized_at.isoformat() if meta.finalized_at else None,
                    json.dumps({k: vars(v) for k, v in meta.fields.items()}),
                ),
            )

    def update_fields(self, wb_id: str, fields: Dict[str, WhiteboardField]) -> None:
        with self._lock, self._conn() as conn:
            row = conn.execute(
                "SELECT fields FROM whiteboards WHERE id=?", (wb_id,)
            ).fetchone()
            if row is None:
                raise KeyError(f"whiteboard {wb_id} not registered")
            existing = json.loads(row[0])
            existing.update({k: vars(v) for k, v in fields.items()})
            conn.execute(
                "UPDATE whiteboards SET fields=? WHERE id=?",
                (json.dumps(existing), wb_id),
            )

    def finalize(self, wb_id: str) -> None:
        with self._lock, self._conn() as conn:
            conn.execute(
                "UPDATE whiteboards SET status=?, finalized_at=? WHERE id=?",
                (
                    WB_STATUS_FINALIZED,
                    datetime.datetime.now(datetime.timezone.utc).isoformat(),
                    wb_id,
                ),
            )

    def get(self, wb_id: str) -> Optional[WhiteboardMeta]:
        with self._lock, self._conn() as conn:
            row = conn.execute(
                "SELECT * FROM whiteboards WHERE id=?", (wb_id,)
            ).fetchone()
        return self._row_to_meta(row) if row else None

    def query(
        self,
        name: Optional[str] = None,
        tags: Sequence[str] = (),
        not_before: Optional[datetime.datetime] = None,
        not_after: Optional[datetime.datetime] = None,
    ) -> Iterable[WhiteboardMeta]:
        sql = "SELECT * FROM whiteboards WHERE 1=1"
        params: List[Any] = []
        if name is not None:
            sql += " AND name=?"
            params.append(name)
        if not_before is not None:
            sql += " AND created_at>=?"
            params.append(not_before.isoformat())
        if not_after is not None:
            sql += " AND created_at<=?"
            params.append(not_after.isoformat())
        sql += " ORDER BY created_at DESC"
        with self._lock, self._conn() as conn:
            rows = conn.execute(sql, params).fetchall()
        metas = [self._row_to_meta(r) for r in rows]
        if tags:
            want = set(tags)
            metas = [m for m in metas if want.issubset(set(m.tags))]
        return metas

    @staticmethod
    def _row_to_meta(row) -> WhiteboardMeta:
        fields = {
            k: WhiteboardField(**v) for k, v in json.loads(row[8]).items()
        }
        return WhiteboardMeta(
            id=row[0],
            name=row[1],
            tags=json.loads(row[2]),
            status=row[3],
            storage_uri=row[4],
            storage_name=row[5],
            created_at=datetime.datetime.fromisoformat(row[6]),
            finalized_at=datetime.datetime.fromisoformat(row[7]) if row[7] else None,
            fields=fields,
        )
