"""Self-contained control-plane store: serving sessions + model registry.

The reference persists all serving-session state on a ClearML server Task
(config objects ``endpoints``/``canary``/``model_monitoring``/``metric_logging``
/``model_monitoring_eps`` plus ``General/*`` parameters, reference:
model_request_processor.py:741-760, 866-867) and resolves models through the
ClearML model repository (Model.query_models). This environment is a single
node with no external server, so the same state machine is backed by a local
SQLite database:

- every config write bumps a per-session ``revision``; readers use it for the
  cheap no-op-when-unchanged check the reference implements with a config hash
  (model_request_processor.py:636-654).
- the model registry supports register/query by project/name-regex/tags/
  published, newest-first -- the exact query surface auto-update needs
  (model_request_processor.py:878-885).
- preprocess code is stored as content-addressed artifacts (sha256), matching
  the reference's artifact upload + hash-verified download
  (preprocess_service.py:63-82).

Multiple serving processes on one node share the store through the filesystem;
SQLite's WAL mode gives cross-process consistency.
"""

import json
import os
import shutil
import sqlite3
import threading
import time
import uuid
from hashlib import sha256
from typing import Any, Dict, List, Optional

from .schemas import ModelRecord

DEFAULT_STORE_ENV = "CLEARML_SERVING_AMD_STORE"
_DEFAULT_STORE_DIR = os.path.expanduser("~/.clearml_serving_amd")

_SCHEMA = """
CREATE TABLE IF NOT EXISTS sessions (
    session_id TEXT PRIMARY KEY,
    name TEXT NOT NULL,
    project TEXT DEFAULT '',
    tags TEXT DEFAULT '[]',
    created REAL,
    revision INTEGER DEFAULT 0
);
CREATE TABLE IF NOT EXISTS config_objects (
    session_id TEXT NOT NULL,
    name TEXT NOT NULL,
    value TEXT NOT NULL,
    PRIMARY KEY (session_id, name)
);
CREATE TABLE IF NOT EXISTS params (
    session_id TEXT NOT NULL,
    name TEXT NOT NULL,
    value TEXT,
    PRIMARY KEY (session_id, name)
);
CREATE TABLE IF NOT EXISTS instances (
    session_id TEXT NOT NULL,
    instance_id TEXT NOT NULL,
    info TEXT DEFAULT '{}',
    last_ping REAL,
    PRIMARY KEY (session_id, instance_id)
);
CREATE TABLE IF NOT EXISTS models (
    model_id TEXT PRIMARY KEY,
    name TEXT NOT NULL,
    project TEXT DEFAULT '',
    tags TEXT DEFAULT '[]',
    framework TEXT,
    uri TEXT,
    published INTEGER DEFAULT 0,
    created REAL
);
CREATE TABLE IF NOT EXISTS artifacts (
    session_id TEXT NOT NULL,
    name TEXT NOT NULL,
    sha256 TEXT NOT NULL,
    path TEXT NOT NULL,
    PRIMARY KEY (session_id, name)
);
"""


class ServingStore:
    """SQLite-backed store; one instance per process, safe across threads."""

    def __init__(self, root: Optional[str] = None):
        self.root = os.path.abspath(
            root or os.environ.get(DEFAULT_STORE_ENV) or _DEFAULT_STORE_DIR
        )
        os.makedirs(self.root, exist_ok=True)
        os.makedirs(os.path.join(self.root, "artifacts"), exist_ok=True)
        os.makedirs(os.path.join(self.root, "models"), exist_ok=True)
        self._db_path = os.path.join(self.root, "store.db")
        self._lock = threading.Lock()
        self._conn = sqlite3.connect(
            self._db_path, check_same_thread=False, timeout=30.0
        )
        self._conn.execute("PRAGMA journal_mode=WAL")
        self._conn.executescript(_SCHEMA)
        self._conn.commit()

    # ------------------------------------------------------------------ #
    # sessions (the "control plane Task" replacement)
    # ------------------------------------------------------------------ #
    def create_session(
        self, name: str = "Serving-Service", project: str = "DevOps",
        tags: Optional[List[str]] = None,
    ) -> str:
        session_id = uuid.uuid4().hex
        with self._lock:
            self._conn.execute(
                "INSERT INTO sessions (session_id, name, project, tags, created)"
                " VALUES (?,?,?,?,?)",
                (session_id, name, project, json.dumps(tags or []), time.time()),
            )
            self._conn.commit()
        return session_id

    def list_sessions(self) -> List[Dict[str, Any]]:
        cur = self._conn.execute(
            "SELECT session_id, name, project, tags, created, revision"
            " FROM sessions ORDER BY created"
        )
        return [
            dict(
                session_id=r[0], name=r[1], project=r[2],
                tags=json.loads(r[3]), created=r[4], revision=r[5],
            )
            for r in cur.fetchall()
        ]

    def get_session(self, session_id: str) -> Optional[Dict[str, Any]]:
        cur = self._conn.execute(
            "SELECT session_id, name, project, tags, created, revision"
            " FROM sessions WHERE session_id=?",
            (session_id,),
        )
        r = cur.fetchone()
        if not r:
            return None
        return dict(
            session_id=r[0], name=r[1], project=r[2], tags=json.loads(r[3]),
            created=r[4], revision=r[5],
        )

    def resolve_session(self, session_id: Optional[str]) -> str:
        """Return session_id, or the first active session when not given
        (mirrors the CLI's 'selecting the first active service' behavior)."""
        if session_id:
            if not self.get_session(session_id):
                raise ValueError("serving session '{}' not found".format(session_id))
            return session_id
        sessions = self.list_sessions()
        if not sessions:
            raise ValueError(
                "no serving session found - create one with 'create' first"
            )
        return sessions[0]["session_id"]

    def revision(self, session_id: str) -> int:
        cur = self._conn.execute(
            "SELECT revision FROM sessions WHERE session_id=?", (session_id,)
        )
        r = cur.fetchone()
        if not r:
            raise ValueError("serving session '{}' not found".format(session_id))
        return int(r[0])

    def _bump(self, session_id: str) -> None:
        self._conn.execute(
            "UPDATE sessions SET revision = revision + 1 WHERE session_id=?",
            (session_id,),
        )

    # ------------------------------------------------------------------ #
    # config objects + params
    # ------------------------------------------------------------------ #
    def set_config_object(self, session_id: str, name: str, value: Any) -> None:
        with self._lock:
            self._conn.execute(
                "INSERT INTO config_objects (session_id, name, value)"
                " VALUES (?,?,?) ON CONFLICT(session_id, name)"
                " DO UPDATE SET value=excluded.value",
                (session_id, name, json.dumps(value)),
            )
            self._bump(session_id)
            self._conn.commit()

    def get_config_object(self, session_id: str, name: str, default=None) -> Any:
        cur = self._conn.execute(
            "SELECT value FROM config_objects WHERE session_id=? AND name=?",
            (session_id, name),
        )
        r = cur.fetchone()
        return json.loads(r[0]) if r else default

    def set_params(self, session_id: str, params: Dict[str, Any]) -> None:
        with self._lock:
            for k, v in params.items():
                self._conn.execute(
                    "INSERT INTO params (session_id, name, value) VALUES (?,?,?)"
                    " ON CONFLICT(session_id, name) DO UPDATE SET value=excluded.value",
                    (session_id, k, json.dumps(v)),
                )
            self._bump(session_id)
            self._conn.commit()

    def ping_instance(self, session_id: str, instance_id: str,
                      info: Optional[Dict[str, Any]] = None) -> None:
        """Serving-container keep-alive (reference parity: the sync daemon
        pings the control Task, model_request_processor.py:999-1007).
        Deliberately does NOT bump the session revision -- pings must not
        look like config changes to other instances."""
        with self._lock:
            self._conn.execute(
                "INSERT INTO instances (session_id, instance_id, info,"
                " last_ping) VALUES (?,?,?,?)"
                " ON CONFLICT(session_id, instance_id)"
                " DO UPDATE SET last_ping=excluded.last_ping,"
                " info=excluded.info",
                (session_id, instance_id, json.dumps(info or {}),
                 time.time()),
            )
            self._conn.commit()

    def list_instances(self, session_id: str,
                       max_age_sec: Optional[float] = None
                       ) -> List[Dict[str, Any]]:
        cur = self._conn.execute(
            "SELECT instance_id, info, last_ping FROM instances"
            " WHERE session_id=? ORDER BY last_ping DESC", (session_id,))
        out = [{"instance_id": r[0], "info": json.loads(r[1] or "{}"),
                "last_ping": r[2]} for r in cur.fetchall()]
        if max_age_sec is not None:
            now = time.time()
            out = [i for i in out if now - (i["last_ping"] or 0) <= max_age_sec]
        return out

    def get_params(self, session_id: str) -> Dict[str, Any]:
        cur = self._conn.execute(
            "SELECT name, value FROM params WHERE session_id=?", (session_id,)
        )
        return {r[0]: json.loads(r[1]) for r in cur.fetchall()}

    # ------------------------------------------------------------------ #
    # model registry
    # ------------------------------------------------------------------ #
    def register_model(
        self, name: str, project: str = "", tags: Optional[List[str]] = None,
        framework: Optional[str] = None, path: Optional[str] = None,
        uri: Optional[str] = None, published: bool = False,
        model_id: Optional[str] = None,
    ) -> ModelRecord:
        """Register a model; when ``path`` is given the file/folder is copied
        into the store (the reference uploads to the ClearML fileserver,
        __main__.py:123-167)."""
        model_id = model_id or uuid.uuid4().hex
        if path:
            dest = os.path.join(self.root, "models", model_id)
            os.makedirs(dest, exist_ok=True)
            if os.path.isdir(path):
                dest = os.path.join(dest, os.path.basename(os.path.normpath(path)))
                shutil.copytree(path, dest, dirs_exist_ok=True)
            else:
                dest = os.path.join(dest, os.path.basename(path))
                shutil.copy2(path, dest)
            uri = dest
        rec = ModelRecord(
            model_id=model_id, name=name, project=project, tags=list(tags or []),
            framework=framework, uri=uri, published=bool(published),
            created=time.time(),
        )
        with self._lock:
            self._conn.execute(
                "INSERT OR REPLACE INTO models"
                " (model_id, name, project, tags, framework, uri, published, created)"
                " VALUES (?,?,?,?,?,?,?,?)",
                (rec.model_id, rec.name, rec.project, json.dumps(rec.tags),
                 rec.framework, rec.uri, int(rec.published), rec.created),
            )
            self._conn.commit()
        return rec

    def get_model(self, model_id: str) -> Optional[ModelRecord]:
        cur = self._conn.execute(
            "SELECT model_id, name, project, tags, framework, uri, published,"
            " created FROM models WHERE model_id=?",
            (model_id,),
        )
        r = cur.fetchone()
        return self._model_row(r) if r else None

    def query_models(
        self, project: Optional[str] = None, name: Optional[str] = None,
        tags: Optional[List[str]] = None, only_published: bool = False,
        max_results: Optional[int] = None,
    ) -> List[ModelRecord]:
        """Newest-first model query; ``name`` is a regex (reference uses
        regexp model-name selection, __main__.py model auto-update --name)."""
        import re

        cur = self._conn.execute(
            "SELECT model_id, name, project, tags, framework, uri, published,"
            " created FROM models ORDER BY created DESC"
        )
        out: List[ModelRecord] = []
        for r in cur.fetchall():
            rec = self._model_row(r)
            if project is not None and rec.project != project:
                continue
            if name is not None and not re.search(name, rec.name):
                continue
            if tags and not set(tags).issubset(set(rec.tags)):
                continue
            if only_published and not rec.published:
                continue
            out.append(rec)
            if max_results and len(out) >= max_results:
                break
        return out

    @staticmethod
    def _model_row(r) -> ModelRecord:
        return ModelRecord(
            model_id=r[0], name=r[1], project=r[2], tags=json.loads(r[3]),
            framework=r[4], uri=r[5], published=bool(r[6]), created=r[7],
        )

    def get_model_local_path(self, model_id: str) -> Optional[str]:
        rec = self.get_model(model_id)
        if not rec or not rec.uri:
            return None
        uri = rec.uri
        if uri.startswith("file://"):
            uri = uri[len("file://"):]
        return uri

    # ------------------------------------------------------------------ #
    # preprocess-code artifacts (content addressed)
    # ------------------------------------------------------------------ #
    def upload_artifact(self, session_id: str, name: str, path: str) -> str:
        """Store a preprocess code file/folder; returns its sha256 digest.

        Folders are zipped (the reference uploads folders as packages,
        preprocess_service.py:84-98 loads either)."""
        if os.path.isdir(path):
            base = os.path.join(self.root, "artifacts", uuid.uuid4().hex)
            archive = shutil.make_archive(base, "zip", path)
            src = archive
        else:
            src = path
        with open(src, "rb") as f:
            digest = sha256(f.read()).hexdigest()
        dest = os.path.join(
            self.root, "artifacts", digest + os.path.splitext(src)[1]
        )
        if not os.path.exists(dest):
            shutil.copy2(src, dest)
        with self._lock:
            self._conn.execute(
                "INSERT OR REPLACE INTO artifacts (session_id, name, sha256, path)"
                " VALUES (?,?,?,?)",
                (session_id, name, digest, dest),
            )
            self._bump(session_id)
            self._conn.commit()
        return digest

    def get_artifact(self, session_id: str, name: str) -> Optional[Dict[str, str]]:
        cur = self._conn.execute(
            "SELECT sha256, path FROM artifacts WHERE session_id=? AND name=?",
            (session_id, name),
        )
        r = cur.fetchone()
        if not r:
            return None
        return {"sha256": r[0], "path": r[1]}

    def close(self) -> None:
        with self._lock:
            self._conn.close()
