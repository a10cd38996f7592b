"""Durable job queue: the RabbitMQ contract of the reference rebuilt as an
embedded multiprocess-safe queue.

Contract preserved from the reference (/root/reference/demo/sender.py:10-31,
worker.py:664-673,650,653-655):
  - durable queue named ``vilbert_multitask_queue``: messages survive process
    crashes (persistent delivery / delivery_mode=2)
  - message payload: JSON {"image_path": [...], "question": str,
    "socket_id": str, "task_id": str}
  - at-least-once: a message is redelivered if the consumer dies before ack
  - manual ack after successful processing only
Improvement over the reference (poison messages redeliver
forever there): a delivery-attempt cap moves messages to a dead-letter table.

Implementation: sqlite3 in WAL mode — multiprocess-safe, durable, zero
external services (offline image has no RabbitMQ). Scale-out keeps the same
competing-consumers model: N workers polling one queue.
"""

from __future__ import annotations

import json
import sqlite3
import time
import uuid
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

QUEUE_NAME = "vilbert_multitask_queue"

_SCHEMA = """
CREATE TABLE IF NOT EXISTS messages (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    queue TEXT NOT NULL,
    body TEXT NOT NULL,
    state TEXT NOT NULL DEFAULT 'ready',   -- ready | unacked | dead
    attempts INTEGER NOT NULL DEFAULT 0,
    consumer TEXT,
    enqueued_at REAL NOT NULL,
    leased_at REAL
);
CREATE INDEX IF NOT EXISTS idx_messages_ready ON messages(queue, state, id);
"""


@dataclass
class Delivery:
    msg_id: int
    body: Dict[str, Any]
    attempts: int


class Broker:
    def __init__(
        self,
        path: str = "vilbert_queue.sqlite3",
        lease_timeout_s: float = 300.0,
        max_attempts: int = 5,
    ):
        self.path = path
        self.lease_timeout_s = lease_timeout_s
        self.max_attempts = max_attempts
        self.consumer_tag = uuid.uuid4().hex[:12]
        self._conn = sqlite3.connect(path, timeout=30.0, check_same_thread=False)
        self._conn.row_factory = sqlite3.Row
        self._conn.execute("PRAGMA journal_mode=WAL")
        # WAL + NORMAL: fsync on checkpoint, not per-commit. Process-crash
        # durability is unchanged (the at-least-once contract's failure
        # model); only whole-machine power loss can drop the tail of the
        # queue. The mixed-task serving config is otherwise fsync-bound on
        # slow disks (measured 2.0k-2.9k req/s tracking the box's disk).
        # VILBERT_SQLITE_FULL_SYNC=1 restores FULL.
        import os as _os

        if _os.environ.get("VILBERT_SQLITE_FULL_SYNC") != "1":
            self._conn.execute("PRAGMA synchronous=NORMAL")
        self._conn.executescript(_SCHEMA)
        self._conn.commit()

    # ---- producer (sender.py:10-31 equivalent) ---------------------------
    def publish(self, body: Dict[str, Any], queue: str = QUEUE_NAME) -> int:
        with self._conn:
            cur = self._conn.execute(
                "INSERT INTO messages (queue, body, enqueued_at) VALUES (?,?,?)",
                (queue, json.dumps(body), time.time()),
            )
            return int(cur.lastrowid)

    # ---- consumer --------------------------------------------------------
    def _requeue_expired(self, queue: str) -> None:
        cutoff = time.time() - self.lease_timeout_s
        with self._conn:
            self._conn.execute(
                "UPDATE messages SET state='ready', consumer=NULL WHERE queue=?"
                " AND state='unacked' AND leased_at < ?",
                (queue, cutoff),
            )

    def get(self, queue: str = QUEUE_NAME, max_n: int = 1) -> List[Delivery]:
        """Lease up to max_n ready messages (at-least-once semantics).

        The claim runs inside BEGIN IMMEDIATE: a deferred transaction that
        reads then upgrades to a write can hit SQLITE_BUSY_SNAPSHOT
        ("database is locked", NOT retried by the busy timeout) when a
        competing consumer commits between the SELECT and the UPDATE —
        taking the reserved lock up front makes contention wait on the
        30 s busy timeout instead of raising.
        """
        self._requeue_expired(queue)
        out: List[Delivery] = []
        try:
            self._conn.execute("BEGIN IMMEDIATE")
        except sqlite3.OperationalError:
            return out  # writer contention past the busy timeout: back off
        try:
            rows = self._conn.execute(
                "SELECT id, body, attempts FROM messages WHERE queue=? AND"
                " state='ready' ORDER BY id LIMIT ?",
                (queue, max_n),
            ).fetchall()
            for r in rows:
                if r["attempts"] >= self.max_attempts:
                    self._conn.execute(
                        "UPDATE messages SET state='dead' WHERE id=?", (r["id"],)
                    )
                    continue
                # atomic claim: the state='ready' guard makes competing
                # consumers safe — SQLite serializes the writes, the loser's
                # UPDATE matches 0 rows and the message is delivered once
                cur = self._conn.execute(
                    "UPDATE messages SET state='unacked', consumer=?, leased_at=?,"
                    " attempts=attempts+1 WHERE id=? AND state='ready'",
                    (self.consumer_tag, time.time(), r["id"]),
                )
                if cur.rowcount == 0:
                    continue  # another consumer won this message
                out.append(Delivery(int(r["id"]), json.loads(r["body"]), int(r["attempts"]) + 1))
            self._conn.execute("COMMIT")
        except BaseException:
            self._conn.execute("ROLLBACK")
            raise
        return out

    def ack(self, msg_id: int) -> None:
        with self._conn:
            self._conn.execute("DELETE FROM messages WHERE id=?", (msg_id,))

    def ack_many(self, msg_ids: List[int]) -> None:
        with self._conn:
            self._conn.executemany(
                "DELETE FROM messages WHERE id=?", [(i,) for i in msg_ids]
            )

    def nack(self, msg_id: int) -> None:
        """Return a message to ready (reference behavior: unacked messages
        redeliver — worker.py:653-655)."""
        with self._conn:
            self._conn.execute(
                "UPDATE messages SET state='ready', consumer=NULL WHERE id=?",
                (msg_id,),
            )

    # ---- introspection ---------------------------------------------------
    def depth(self, queue: str = QUEUE_NAME) -> int:
        return self._conn.execute(
            "SELECT COUNT(*) FROM messages WHERE queue=? AND state='ready'", (queue,)
        ).fetchone()[0]

    def dead_count(self, queue: str = QUEUE_NAME) -> int:
        return self._conn.execute(
            "SELECT COUNT(*) FROM messages WHERE queue=? AND state='dead'", (queue,)
        ).fetchone()[0]

    def close(self) -> None:
        self._conn.close()


def vilbert_task(
    broker: Broker,
    image_paths: List[str],
    question: str,
    task_id: int,
    socket_id: str,
    trace_id: Optional[str] = None,
) -> int:
    """Producer helper mirroring demo/sender.py:10-31's message schema
    (+ an optional trace_id the reference lacks — utils/trace.py)."""
    body: Dict[str, Any] = {
        "image_path": image_paths,
        "question": question,
        "socket_id": socket_id,
        "task_id": str(task_id),
    }
    if trace_id:
        body["trace_id"] = trace_id
    return broker.publish(body)
