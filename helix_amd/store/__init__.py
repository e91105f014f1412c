"""Persistence layer — SQLite-backed document store.

The reference uses Postgres/GORM with ~75 tables plus an in-memory twin
for tests (SURVEY.md §2.1 Store, §4 in-memory fakes). Here one Store
serves both roles: file-backed SQLite for `serve`, ":memory:" for tests.
Rows are JSON documents with indexed hot columns.
"""
from __future__ import annotations

import json
import sqlite3
import threading
import time
from typing import Any, Dict, List, Optional

_TABLES = [
    "users", "api_keys", "apps", "sessions", "interactions", "llm_calls",
    "usage_metrics", "knowledge", "knowledge_versions", "models",
    "provider_endpoints", "secrets", "step_info", "memories", "oauth_tokens",
    "triggers", "organizations", "teams", "memberships", "access_grants",
    "runner_profiles", "projects", "spec_tasks", "git_repositories",
    "evaluation_runs", "system_settings", "wallets", "transactions",
    "rag_chunks", "usage_rollups", "runner_assignments",
    "org_positions", "org_bots", "org_streams", "org_messages",
]


class Store:
    def __init__(self, path: str = ":memory:"):
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.execute("PRAGMA journal_mode=WAL")
        self._lock = threading.RLock()
        with self._lock:
            for t in _TABLES:
                self._db.execute(
                    f"""CREATE TABLE IF NOT EXISTS {t} (
                        id TEXT PRIMARY KEY,
                        owner TEXT DEFAULT '',
                        parent TEXT DEFAULT '',
                        created REAL DEFAULT 0,
                        updated REAL DEFAULT 0,
                        doc TEXT NOT NULL
                    )""")
                self._db.execute(
                    f"CREATE INDEX IF NOT EXISTS idx_{t}_owner ON {t}(owner)")
                self._db.execute(
                    f"CREATE INDEX IF NOT EXISTS idx_{t}_parent ON {t}(parent)")
            self._db.commit()

    # ------------------------------------------------------------------
    def put(self, table: str, id: str, doc: Dict[str, Any],
            owner: str = "", parent: str = ""):
        now = time.time()
        blob = json.dumps(doc, default=str)
        with self._lock:
            self._db.execute(
                f"""INSERT INTO {table} (id, owner, parent, created, updated, doc)
                    VALUES (?,?,?,?,?,?)
                    ON CONFLICT(id) DO UPDATE SET
                      owner=excluded.owner, parent=excluded.parent,
                      updated=excluded.updated, doc=excluded.doc""",
                (id, owner, parent, now, now, blob))
            self._db.commit()

    def backup(self, path: str):
        """Consistent online backup (sqlite3 backup API) — the
        operational role of the reference's Postgres dump/restore."""
        import sqlite3 as _sq
        with self._lock:
            dst = _sq.connect(path)
            try:
                self._db.backup(dst)
            finally:
                dst.close()

    def get(self, table: str, id: str) -> Optional[Dict[str, Any]]:
        with self._lock:
            row = self._db.execute(
                f"SELECT doc FROM {table} WHERE id=?", (id,)).fetchone()
        return json.loads(row[0]) if row else None

    def delete(self, table: str, id: str) -> bool:
        with self._lock:
            cur = self._db.execute(f"DELETE FROM {table} WHERE id=?", (id,))
            self._db.commit()
            return cur.rowcount > 0

    def list(self, table: str, owner: Optional[str] = None,
             parent: Optional[str] = None, limit: int = 1000,
             offset: int = 0, desc: bool = True) -> List[Dict[str, Any]]:
        q = f"SELECT doc FROM {table}"
        conds, args = [], []
        if owner is not None:
            conds.append("owner=?")
            args.append(owner)
        if parent is not None:
            conds.append("parent=?")
            args.append(parent)
        if conds:
            q += " WHERE " + " AND ".join(conds)
        q += f" ORDER BY created {'DESC' if desc else 'ASC'} LIMIT ? OFFSET ?"
        args.extend([limit, offset])
        with self._lock:
            rows = self._db.execute(q, args).fetchall()
        return [json.loads(r[0]) for r in rows]

    def count(self, table: str, owner: Optional[str] = None) -> int:
        q = f"SELECT COUNT(*) FROM {table}"
        args = []
        if owner is not None:
            q += " WHERE owner=?"
            args.append(owner)
        with self._lock:
            return self._db.execute(q, args).fetchone()[0]

    def find_one(self, table: str, **fields) -> Optional[Dict[str, Any]]:
        """Linear scan matching on doc fields (small tables only)."""
        for doc in self.list(table, limit=100000):
            if all(doc.get(k) == v for k, v in fields.items()):
                return doc
        return None

    def close(self):
        with self._lock:
            self._db.close()
