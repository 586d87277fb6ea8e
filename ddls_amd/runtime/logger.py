"""Result logging: background-thread save of nested result dicts to
gzip-pickle or a sqlite key-value store.

Reference: ``ddls/loggers/logger.py:11-87`` (which used SqliteDict; this
rebuild uses a small stdlib-sqlite3 KV table with the same append semantics).
"""
from __future__ import annotations

import gzip
import os
import pickle
import sqlite3
import threading
from typing import Dict, Optional


class SqliteKV:
    """Minimal SqliteDict-alike: pickle-valued key/value table."""

    def __init__(self, path: str):
        self.path = path
        self._conn = sqlite3.connect(path)
        self._conn.execute(
            "CREATE TABLE IF NOT EXISTS kv (key TEXT PRIMARY KEY, value BLOB)")

    def __getitem__(self, key):
        row = self._conn.execute("SELECT value FROM kv WHERE key=?",
                                 (key,)).fetchone()
        if row is None:
            raise KeyError(key)
        return pickle.loads(row[0])

    def get(self, key, default=None):
        try:
            return self[key]
        except KeyError:
            return default

    def __contains__(self, key):
        return self._conn.execute("SELECT 1 FROM kv WHERE key=?",
                                  (key,)).fetchone() is not None

    def __setitem__(self, key, value):
        self._conn.execute(
            "INSERT OR REPLACE INTO kv (key, value) VALUES (?, ?)",
            (key, pickle.dumps(value)))

    def keys(self):
        return [r[0] for r in self._conn.execute("SELECT key FROM kv")]

    def commit(self):
        self._conn.commit()

    def close(self):
        self._conn.close()


class Logger:
    """Append-style results logger with a background save thread
    (reference semantics: list values are extended across writes)."""

    def __init__(self, path_to_save: str, name: str = "results",
                 use_sqlite_database: bool = False):
        self.path_to_save = path_to_save
        self.name = name
        self.use_sqlite_database = use_sqlite_database
        os.makedirs(path_to_save, exist_ok=True)
        self._thread: Optional[threading.Thread] = None

    def write(self, results: Dict[str, dict], block: bool = False):
        self.join()
        self._thread = threading.Thread(target=self._save, args=(results,))
        self._thread.start()
        if block:
            self.join()

    def join(self):
        if self._thread is not None:
            self._thread.join()
            self._thread = None

    def _save(self, results: Dict[str, dict]):
        for log_name, log in results.items():
            path = os.path.join(self.path_to_save, log_name)
            if self.use_sqlite_database:
                db = SqliteKV(path + ".sqlite")
                for key, val in log.items():
                    if key in db and isinstance(val, list):
                        db[key] = db[key] + val
                    else:
                        db[key] = val
                db.commit()
                db.close()
            else:
                existing = {}
                if os.path.exists(path + ".pkl"):
                    with gzip.open(path + ".pkl", "rb") as f:
                        existing = pickle.load(f)
                for key, val in log.items():
                    if key in existing and isinstance(val, list):
                        existing[key] = existing[key] + val
                    else:
                        existing[key] = val
                with gzip.open(path + ".pkl", "wb") as f:
                    pickle.dump(existing, f)

    def load(self, log_name: str) -> dict:
        path = os.path.join(self.path_to_save, log_name)
        if self.use_sqlite_database:
            db = SqliteKV(path + ".sqlite")
            out = {k: db[k] for k in db.keys()}
            db.close()
            return out
        with gzip.open(path + ".pkl", "rb") as f:
            return pickle.load(f)


class ExternalLogger:
    """Switchable external metric logger seam (the reference pushes every
    epoch's unpacked results to wandb, ``rllib_epoch_loop.py:144-230``; this
    rebuild logs to local sqlite/pkl tables by default and mirrors to any
    wandb-API-compatible backend when one is attached).

    Backend contract: an object with ``log(metrics: dict, step: int)`` and
    optionally ``finish()``.  ``ExternalLogger.wandb(...)`` builds one from
    the wandb package if it is importable (no-op otherwise — this image has
    no network, so the seam is exercised with fakes in tests).
    """

    def __init__(self, backend=None):
        self.backend = backend

    @classmethod
    def wandb(cls, project: str = "ddls_amd", **init_kwargs):
        try:
            import wandb  # noqa: F401
        except Exception:
            return cls(backend=None)
        run = wandb.init(project=project, **init_kwargs)
        return cls(backend=run)

    def log(self, metrics: Dict, step: Optional[int] = None):
        if self.backend is None:
            return
        scalars = {k: v for k, v in metrics.items()
                   if isinstance(v, (int, float))}
        self.backend.log(scalars, step=step)

    def finish(self):
        if self.backend is not None and hasattr(self.backend, "finish"):
            self.backend.finish()
