"""Persistence layer — SQLite-backed document store.

The reference uses Postgres/GORM with ~75 tables plus an in-memory twin
for tests (SURVEY.md §2.1 Store, §4 in-memory fakes). Here one Store
serves both roles: file-backed SQLite for `serve`, ":memory:" for tests.
Rows are JSON documents with indexed hot columns.

Concurrency design (round 2):
- file-backed DBs run WAL with ONE writer connection (short write lock)
  and per-thread READER connections, so streaming sessions reading
  sessions/interactions never queue behind llm_call inserts;
- hot high-frequency writers (llm_calls, step_info, usage_metrics,
  interaction partial-persist) can pass ``buffered=True``: writes land
  in an in-memory accumulator flushed as one transaction every 200 ms —
  the reference's websocket-streaming accumulator behavior
  (design/2026-02-25, wsprotocol/accumulator.go:15: DB writes -75%).
  ``get`` consults the buffer first; ``list``/``count``/``find_one``
  flush the table before querying, so readers never see stale windows.
- ":memory:" keeps the single-connection + RLock layout (separate
  sqlite connections do not share a memory DB); tests exercise the same
  buffered API through it.
"""
from __future__ import annotations

import json
import sqlite3
import threading
import time
from typing import Any, Dict, List, Optional, Tuple

_TABLES = [
    "users", "api_keys", "apps", "sessions", "interactions", "llm_calls",
    "usage_metrics", "knowledge", "knowledge_versions", "models",
    "provider_endpoints", "secrets", "step_info", "memories", "oauth_tokens",
    "triggers", "organizations", "teams", "memberships", "access_grants",
    "runner_profiles", "projects", "spec_tasks", "git_repositories",
    "evaluation_runs", "system_settings", "wallets", "transactions",
    "rag_chunks", "rag_alias", "oidc_states", "billing_customers", "billing_events", "billing_invoices", "sandboxes", "bus_messages", "bus_consumers", "error_events", "usage_rollups", "runner_assignments",
    "org_positions", "org_bots", "org_streams", "org_messages", "org_audit",
]

_FLUSH_INTERVAL = 0.2   # reference accumulator cadence (200 ms)


class Store:
    def __init__(self, path: str = ":memory:",
                 flush_interval: float = _FLUSH_INTERVAL):
        self._path = path
        self._memory = path == ":memory:"
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.execute("PRAGMA journal_mode=WAL")
        self._db.execute("PRAGMA synchronous=NORMAL")
        self._lock = threading.RLock()       # writer lock
        self._local = threading.local()      # per-thread reader conns
        self._closed = False
        # write accumulator: (table, id) -> (owner, parent, doc_json)
        self._pending: Dict[Tuple[str, str], Tuple[str, str, str, float]] = {}
        self._pending_lock = threading.Lock()
        self._flush_interval = flush_interval
        self._flusher: Optional[threading.Thread] = None
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

    # -- connections ----------------------------------------------------
    def _reader(self) -> sqlite3.Connection:
        """Per-thread read connection (WAL => readers don't block on the
        writer). Memory DBs fall back to the shared connection."""
        if self._memory:
            return self._db
        conn = getattr(self._local, "conn", None)
        if conn is None:
            conn = sqlite3.connect(self._path, check_same_thread=False)
            conn.execute("PRAGMA query_only=1")
            self._local.conn = conn
        return conn

    def _read(self, query: str, args) -> list:
        if self._memory:
            with self._lock:
                return self._db.execute(query, args).fetchall()
        return self._reader().execute(query, args).fetchall()

    # -- accumulator ----------------------------------------------------
    def _ensure_flusher(self):
        if self._flusher is None or not self._flusher.is_alive():
            self._flusher = threading.Thread(target=self._flush_loop,
                                             daemon=True,
                                             name="store-flusher")
            self._flusher.start()

    def _flush_loop(self):
        while not self._closed:
            time.sleep(self._flush_interval)
            try:
                self.flush()
            except Exception:
                pass

    def flush(self, table: Optional[str] = None):
        """Commit pending buffered writes (all tables or one) in a single
        transaction. Last-write-wins per id, like the reference's
        websocket accumulator. Entries stay in the buffer until AFTER
        the commit — a concurrent get() must hit either the buffer or
        the committed row, never the gap between them (WAL readers do
        not see uncommitted writes)."""
        with self._pending_lock:
            if table is None:
                items = list(self._pending.items())
            else:
                items = [(k, v) for k, v in self._pending.items()
                         if k[0] == table]
        if not items:
            return
        with self._lock:
            for (tbl, id_), (owner, parent, blob, ts) in items:
                self._db.execute(
                    f"""INSERT INTO {tbl} (id, owner, parent, created,
                                           updated, doc)
                        VALUES (?,?,?,?,?,?)
                        ON CONFLICT(id) DO UPDATE SET
                          owner=excluded.owner, parent=excluded.parent,
                          updated=excluded.updated, doc=excluded.doc""",
                    (id_, owner, parent, ts, ts, blob))
            self._db.commit()
        with self._pending_lock:
            for k, v in items:
                # drop only if not overwritten since the snapshot
                if self._pending.get(k) is v:
                    del self._pending[k]

    # ------------------------------------------------------------------
    def put(self, table: str, id: str, doc: Dict[str, Any],
            owner: str = "", parent: str = "", buffered: bool = False):
        blob = json.dumps(doc, default=str)
        if buffered:
            with self._pending_lock:
                self._pending[(table, id)] = (owner, parent, blob,
                                              time.time())
            self._ensure_flusher()
            return
        now = time.time()
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
        self.flush()
        import sqlite3 as _sq
        with self._lock:
            dst = _sq.connect(path)
            try:
                self._db.backup(dst)
            finally:
                dst.close()

    def get(self, table: str, id: str) -> Optional[Dict[str, Any]]:
        with self._pending_lock:
            hit = self._pending.get((table, id))
        if hit is not None:
            return json.loads(hit[2])
        row = self._read(f"SELECT doc FROM {table} WHERE id=?", (id,))
        return json.loads(row[0][0]) if row else None

    def delete(self, table: str, id: str) -> bool:
        with self._pending_lock:
            self._pending.pop((table, id), None)
        with self._lock:
            cur = self._db.execute(f"DELETE FROM {table} WHERE id=?", (id,))
            self._db.commit()
            return cur.rowcount > 0

    def list(self, table: str, owner: Optional[str] = None,
             parent: Optional[str] = None, limit: int = 1000,
             offset: int = 0, desc: bool = True) -> List[Dict[str, Any]]:
        self.flush(table)
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
        rows = self._read(q, args)
        return [json.loads(r[0]) for r in rows]

    def count(self, table: str, owner: Optional[str] = None) -> int:
        self.flush(table)
        q = f"SELECT COUNT(*) FROM {table}"
        args = []
        if owner is not None:
            q += " WHERE owner=?"
            args.append(owner)
        return self._read(q, args)[0][0]

    def find_one(self, table: str, **fields) -> Optional[Dict[str, Any]]:
        """Linear scan matching on doc fields (small tables only)."""
        for doc in self.list(table, limit=100000):
            if all(doc.get(k) == v for k, v in fields.items()):
                return doc
        return None

    def close(self):
        self._closed = True
        try:
            self.flush()
        except Exception:
            pass
        with self._lock:
            self._db.close()
