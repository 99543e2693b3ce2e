"""Pluggable GCS persistence backends (reference: gcs_server can run
against in-memory storage or external Redis for HA —
gcs/gcs_server/gcs_server.cc store selection, RAY_REDIS_ADDRESS).

There is no network service in this deployment, so the HA-grade
backend is sqlite in WAL mode on shared/durable storage:

- FileStorage  — periodic whole-state snapshots with atomic replace
  (loses at most one persist-loop period on a crash).
- SqliteStorage — the same snapshots PLUS a synchronously-committed
  per-mutation KV journal replayed on top of the latest snapshot, so
  an acknowledged internal-KV write survives kill -9 of the GCS
  process. Snapshot writes compact the journal.

Select with RAY_AMD_GCS_STORAGE=file|sqlite (default file).
"""
from __future__ import annotations

import os
import pickle
import sqlite3
from typing import List, Optional, Tuple


class FileStorage:
    kind = "file"

    def __init__(self, path: str):
        self.path = path

    def save_snapshot(self, blob: bytes) -> None:
        tmp = self.path + ".tmp"
        with open(tmp, "wb") as f:
            f.write(blob)
        os.replace(tmp, self.path)

    def load(self) -> Tuple[Optional[bytes], List[bytes]]:
        if not os.path.exists(self.path):
            return None, []
        try:
            with open(self.path, "rb") as f:
                return f.read(), []
        except Exception:
            return None, []

    def journal(self, op: bytes) -> None:
        pass  # snapshot-only backend

    def close(self) -> None:
        pass


class SqliteStorage:
    kind = "sqlite"

    def __init__(self, path: str):
        self.path = path
        self.db = sqlite3.connect(path)
        self.db.execute("PRAGMA journal_mode=WAL")
        self.db.execute("PRAGMA synchronous=NORMAL")
        self.db.execute(
            "CREATE TABLE IF NOT EXISTS snapshot ("
            "id INTEGER PRIMARY KEY CHECK (id = 1), blob BLOB)")
        self.db.execute(
            "CREATE TABLE IF NOT EXISTS journal ("
            "seq INTEGER PRIMARY KEY AUTOINCREMENT, op BLOB)")
        self.db.commit()

    def save_snapshot(self, blob: bytes) -> None:
        with self.db:  # snapshot + journal compaction, one txn
            self.db.execute(
                "INSERT OR REPLACE INTO snapshot (id, blob) VALUES (1, ?)",
                (blob,))
            self.db.execute("DELETE FROM journal")

    def load(self) -> Tuple[Optional[bytes], List[bytes]]:
        row = self.db.execute(
            "SELECT blob FROM snapshot WHERE id = 1").fetchone()
        ops = [r[0] for r in self.db.execute(
            "SELECT op FROM journal ORDER BY seq")]
        return (row[0] if row else None), ops

    def journal(self, op: bytes) -> None:
        with self.db:
            self.db.execute("INSERT INTO journal (op) VALUES (?)", (op,))

    def close(self) -> None:
        try:
            self.db.close()
        except Exception:
            pass


def open_storage(path: str):
    """Backend factory: RAY_AMD_GCS_STORAGE=sqlite or a .db/.sqlite
    persist path selects sqlite; anything else the file backend."""
    kind = os.environ.get("RAY_AMD_GCS_STORAGE", "").lower()
    if kind == "sqlite" or path.endswith((".db", ".sqlite")):
        if not path.endswith((".db", ".sqlite")):
            path = path + ".db"
        return SqliteStorage(path)
    return FileStorage(path)


def encode_op(*op) -> bytes:
    return pickle.dumps(op)


def decode_op(blob: bytes):
    return pickle.loads(blob)
