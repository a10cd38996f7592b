"""Persistence layer: the reference's Postgres/Django-ORM tables rebuilt on
sqlite3 (offline image has no Postgres server; schema and field names are
behavior-identical to /root/reference/demo/models.py:18-46 so the HTTP
responses match).

Tables:
  tasks           (demo/models.py:18-28, table name "tasks")
  questionanswer  (demo/models.py:31-43, table name "questionanswer")
"""

from __future__ import annotations

import json
import sqlite3
import threading
import time
from typing import Any, Dict, List, Optional

_SCHEMA = """
CREATE TABLE IF NOT EXISTS tasks (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    created TEXT NOT NULL,
    modified TEXT NOT NULL,
    unique_id INTEGER UNIQUE NOT NULL,
    name TEXT NOT NULL,
    placeholder TEXT DEFAULT '',
    description TEXT DEFAULT '',
    num_of_images INTEGER DEFAULT 1,
    example TEXT DEFAULT ''
);
CREATE TABLE IF NOT EXISTS attachment (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    created TEXT NOT NULL,
    modified TEXT NOT NULL,
    attachment TEXT
);
CREATE TABLE IF NOT EXISTS questionanswer (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    created TEXT NOT NULL,
    modified TEXT NOT NULL,
    task_id INTEGER,
    input_text TEXT,
    input_images TEXT,
    answer_text TEXT,
    answer_images TEXT,
    socket_id TEXT
);
"""


def _now() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%S")


class Database:
    def __init__(self, path: str = "vilbert_demo.sqlite3"):
        self.path = path
        self._local = threading.local()
        with self._conn() as c:
            c.executescript(_SCHEMA)

    def _conn(self) -> sqlite3.Connection:
        conn = getattr(self._local, "conn", None)
        if conn is None:
            conn = sqlite3.connect(self.path, timeout=30.0)
            conn.row_factory = sqlite3.Row
            conn.execute("PRAGMA journal_mode=WAL")
            # see broker.py: WAL+NORMAL keeps process-crash durability
            import os as _os

            if _os.environ.get("VILBERT_SQLITE_FULL_SYNC") != "1":
                conn.execute("PRAGMA synchronous=NORMAL")
            self._local.conn = conn
        return conn

    # ---- tasks -----------------------------------------------------------
    def seed_tasks(self) -> None:
        """Populate the task table from the registry (replaces the reference's
        hand-entered admin rows)."""
        from ..tasks import TASKS

        with self._conn() as c:
            for t in TASKS.values():
                c.execute(
                    "INSERT OR IGNORE INTO tasks (created, modified, unique_id, name,"
                    " placeholder, description, num_of_images, example)"
                    " VALUES (?,?,?,?,?,?,?,?)",
                    (
                        _now(), _now(), t.task_id, t.name,
                        f"Ask a question for {t.name}", t.decode.value,
                        t.min_images, "",
                    ),
                )

    def get_task(self, unique_id: int) -> Optional[Dict[str, Any]]:
        row = self._conn().execute(
            "SELECT * FROM tasks WHERE unique_id=?", (unique_id,)
        ).fetchone()
        return dict(row) if row else None

    # ---- questionanswer --------------------------------------------------
    def create_question(
        self, task_id: int, input_text: str, input_images: List[str], socket_id: str
    ) -> int:
        with self._conn() as c:
            cur = c.execute(
                "INSERT INTO questionanswer (created, modified, task_id, input_text,"
                " input_images, socket_id) VALUES (?,?,?,?,?,?)",
                (_now(), _now(), task_id, input_text, json.dumps(input_images), socket_id),
            )
            return int(cur.lastrowid)

    def create_questions(self, rows: List[tuple]) -> List[int]:
        """Batch insert [(task_id, input_text, input_images_json, socket_id)]
        in ONE transaction (the per-row commit was the mixed-serving
        bottleneck: 64 WAL fsyncs per batch)."""
        with self._conn() as c:
            first = c.execute("SELECT COALESCE(MAX(id), 0) FROM questionanswer").fetchone()[0]
            c.executemany(
                "INSERT INTO questionanswer (created, modified, task_id, input_text,"
                " input_images, socket_id) VALUES (?,?,?,?,?,?)",
                [(_now(), _now(), t, q, imgs, sid) for (t, q, imgs, sid) in rows],
            )
            return list(range(first + 1, first + 1 + len(rows)))

    def save_answers(self, items: List[tuple]) -> None:
        """Batch update [(qa_id, answer_text)] in one transaction."""
        with self._conn() as c:
            c.executemany(
                "UPDATE questionanswer SET answer_text=?, modified=? WHERE id=?",
                [(ans, _now(), qa) for (qa, ans) in items],
            )

    def save_answer(
        self, qa_id: int, answer_text: str, answer_images: Optional[List[str]] = None
    ) -> None:
        with self._conn() as c:
            c.execute(
                "UPDATE questionanswer SET answer_text=?, answer_images=?, modified=?"
                " WHERE id=?",
                (answer_text, json.dumps(answer_images or []), _now(), qa_id),
            )

    def add_attachment(self, path: str) -> int:
        """Attachment rows (demo/models.py:45-46): uploaded-file records."""
        with self._conn() as c:
            cur = c.execute(
                "INSERT INTO attachment (created, modified, attachment) VALUES (?,?,?)",
                (_now(), _now(), path),
            )
            return int(cur.lastrowid)

    def recent_questions(self, limit: int = 50) -> List[Dict[str, Any]]:
        rows = self._conn().execute(
            "SELECT * FROM questionanswer ORDER BY id DESC LIMIT ?", (limit,)
        ).fetchall()
        return [dict(r) for r in rows]

    def list_tasks(self) -> List[Dict[str, Any]]:
        rows = self._conn().execute("SELECT * FROM tasks ORDER BY unique_id").fetchall()
        return [dict(r) for r in rows]

    def update_task(self, unique_id: int, fields: Dict[str, Any]) -> bool:
        """Admin CRUD (demo/admin.py equivalent): edit a Tasks row's
        editable columns."""
        allowed = {"name", "placeholder", "description", "num_of_images", "example"}
        cols = {k: v for k, v in fields.items() if k in allowed}
        if not cols:
            return False
        sets = ", ".join(f"{k}=?" for k in cols) + ", modified=?"
        with self._conn() as c:
            cur = c.execute(
                f"UPDATE tasks SET {sets} WHERE unique_id=?",
                [*cols.values(), _now(), unique_id],
            )
            return cur.rowcount > 0

    def delete_question(self, qa_id: int) -> bool:
        with self._conn() as c:
            return c.execute("DELETE FROM questionanswer WHERE id=?", (qa_id,)).rowcount > 0

    def get_question(self, qa_id: int) -> Optional[Dict[str, Any]]:
        row = self._conn().execute(
            "SELECT * FROM questionanswer WHERE id=?", (qa_id,)
        ).fetchone()
        return dict(row) if row else None

    def close(self) -> None:
        conn = getattr(self._local, "conn", None)
        if conn is not None:
            conn.close()
            self._local.conn = None
