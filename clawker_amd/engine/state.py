"""Sandbox registry (sqlite) — the engine's durable view of created
sandboxes; live state comes from each sandbox's rundir status files.
(The reference keeps this in dockerd; its CP keeps an agent registry in
sqlite — registry_sqlite.go. Single-writer discipline: only the engine
process holding the connection writes; WAL mode for concurrent readers.)
"""
from __future__ import annotations

import json
import sqlite3
import threading
import time
from pathlib import Path

from .. import consts

_SCHEMA = """
CREATE TABLE IF NOT EXISTS sandboxes (
  name TEXT PRIMARY KEY,
  project TEXT NOT NULL DEFAULT '',
  agent TEXT NOT NULL DEFAULT '',
  image TEXT NOT NULL DEFAULT '',
  created REAL NOT NULL,
  labels TEXT NOT NULL DEFAULT '{}',
  gpus TEXT NOT NULL DEFAULT '[]',
  spec_path TEXT NOT NULL DEFAULT '',
  rundir TEXT NOT NULL DEFAULT '',
  statedir TEXT NOT NULL DEFAULT ''
);
CREATE TABLE IF NOT EXISTS volumes (
  name TEXT PRIMARY KEY,
  created REAL NOT NULL,
  labels TEXT NOT NULL DEFAULT '{}',
  path TEXT NOT NULL DEFAULT ''
);
"""


class StateDB:
    def __init__(self, path: Path | None = None):
        self.path = path or (consts.data_dir() / "state.db")
        self.path.parent.mkdir(parents=True, exist_ok=True)
        # check_same_thread=False: the CP daemon reads from watcher/admin
        # threads; python's sqlite3 serializes access internally
        self.db = sqlite3.connect(str(self.path), timeout=10.0,
                                  check_same_thread=False)
        # sqlite3 serializes single statements, but an execute+commit
        # PAIR from two threads interleaves ("cannot commit - no
        # transaction is active" under concurrent creates): every
        # mutation holds this lock across its transaction
        self._lock = threading.RLock()
        self.db.execute("PRAGMA journal_mode=WAL")
        self.db.execute("PRAGMA busy_timeout=10000")
        self.db.executescript(_SCHEMA)
        self.db.commit()

    def close(self) -> None:
        self.db.close()

    # -- sandboxes -------------------------------------------------------------
    def add_sandbox(self, name: str, project: str, agent: str, image: str,
                    labels: dict, gpus: list[int], spec_path: str,
                    rundir: str, statedir: str) -> None:
        with self._lock:
            self.db.execute(
                "INSERT OR REPLACE INTO sandboxes "
                "(name,project,agent,image,created,labels,gpus,spec_path,"
                "rundir,statedir) VALUES (?,?,?,?,?,?,?,?,?,?)",
                (name, project, agent, image, time.time(), json.dumps(labels),
                 json.dumps(gpus), spec_path, rundir, statedir))
            self.db.commit()

    def get_sandbox(self, name: str) -> dict | None:
        cur = self.db.execute("SELECT * FROM sandboxes WHERE name=?", (name,))
        row = cur.fetchone()
        if row is None:
            return None
        cols = [d[0] for d in cur.description]
        d = dict(zip(cols, row))
        d["labels"] = json.loads(d["labels"])
        d["gpus"] = json.loads(d["gpus"])
        return d

    def list_sandboxes(self, project: str | None = None,
                       label_filters: dict | None = None) -> list[dict]:
        cur = self.db.execute("SELECT * FROM sandboxes ORDER BY created")
        cols = [d[0] for d in cur.description]
        out = []
        for row in cur.fetchall():
            d = dict(zip(cols, row))
            d["labels"] = json.loads(d["labels"])
            d["gpus"] = json.loads(d["gpus"])
            if project and d["project"] != project:
                continue
            if label_filters and any(d["labels"].get(k) != v for k, v in label_filters.items()):
                continue
            out.append(d)
        return out

    def remove_sandbox(self, name: str) -> None:
        with self._lock:
            self.db.execute("DELETE FROM sandboxes WHERE name=?", (name,))
            self.db.commit()

    # -- volumes ---------------------------------------------------------------
    def add_volume(self, name: str, path: str, labels: dict) -> None:
        with self._lock:
            self.db.execute(
                "INSERT OR REPLACE INTO volumes (name,created,labels,path) "
                "VALUES (?,?,?,?)",
                (name, time.time(), json.dumps(labels), path))
            self.db.commit()

    def get_volume(self, name: str) -> dict | None:
        cur = self.db.execute("SELECT * FROM volumes WHERE name=?", (name,))
        row = cur.fetchone()
        if row is None:
            return None
        cols = [d[0] for d in cur.description]
        d = dict(zip(cols, row))
        d["labels"] = json.loads(d["labels"])
        return d

    def list_volumes(self) -> list[dict]:
        cur = self.db.execute("SELECT * FROM volumes ORDER BY created")
        cols = [d[0] for d in cur.description]
        out = []
        for row in cur.fetchall():
            d = dict(zip(cols, row))
            d["labels"] = json.loads(d["labels"])
            out.append(d)
        return out

    def remove_volume(self, name: str) -> None:
        with self._lock:
            self.db.execute("DELETE FROM volumes WHERE name=?", (name,))
            self.db.commit()
