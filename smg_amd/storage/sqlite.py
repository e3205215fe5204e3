"""SQL-backed Responses/Conversations storage (reference:
crates/data_connector/src/postgres.rs + oracle.rs — durable SQL persistence
behind the same ResponseStorage/ConversationStorage traits).

The image has no postgres/oracle client, so the in-tree SQL driver is
sqlite3 (stdlib): same schema shape (responses, response_input_items,
conversations, conversation_items), same trait surface, WAL mode for
concurrent readers.  `make_storage("sqlite")` uses an in-memory database;
`make_storage("sqlite:///path/to.db")` persists on disk.
"""
from __future__ import annotations

import json
import sqlite3
import threading
import time
import uuid
from typing import Any, Dict, List, Optional

from .base import ConversationStorage, ResponseStorage, StorageError

_SCHEMA = """
CREATE TABLE IF NOT EXISTS responses (
    id TEXT PRIMARY KEY,
    body TEXT NOT NULL,
    created_at INTEGER NOT NULL
);
CREATE TABLE IF NOT EXISTS response_input_items (
    response_id TEXT NOT NULL,
    seq INTEGER NOT NULL,
    body TEXT NOT NULL,
    PRIMARY KEY (response_id, seq)
);
CREATE TABLE IF NOT EXISTS conversations (
    id TEXT PRIMARY KEY,
    metadata TEXT NOT NULL,
    created_at INTEGER NOT NULL
);
CREATE TABLE IF NOT EXISTS conversation_items (
    conversation_id TEXT NOT NULL,
    item_id TEXT NOT NULL,
    seq INTEGER NOT NULL,
    body TEXT NOT NULL,
    PRIMARY KEY (conversation_id, item_id)
);
CREATE INDEX IF NOT EXISTS idx_conv_items_seq
    ON conversation_items (conversation_id, seq);
"""


class _Db:
    """One shared connection + lock; sqlite calls are sub-ms so they run
    inline on the event loop (aiosqlite is not in the image)."""

    def __init__(self, path: str):
        self.conn = sqlite3.connect(path, check_same_thread=False)
        self.conn.row_factory = sqlite3.Row
        self.lock = threading.Lock()
        with self.lock:
            if path != ":memory:":
                self.conn.execute("PRAGMA journal_mode=WAL")
            self.conn.executescript(_SCHEMA)
            self.conn.commit()

    def exec(self, sql: str, params=()) -> sqlite3.Cursor:
        with self.lock:
            cur = self.conn.execute(sql, params)
            self.conn.commit()
            return cur

    def query(self, sql: str, params=()) -> List[sqlite3.Row]:
        with self.lock:
            return self.conn.execute(sql, params).fetchall()


class SqliteResponseStorage(ResponseStorage):
    def __init__(self, db: _Db):
        self._db = db

    async def store_response(self, response: Dict[str, Any]) -> str:
        rid = response.get("id") or f"resp_{uuid.uuid4().hex}"
        response["id"] = rid
        inputs = response.pop("_input_items", [])
        self._db.exec(
            "INSERT OR REPLACE INTO responses (id, body, created_at) VALUES (?, ?, ?)",
            (rid, json.dumps(response), response.get("created_at") or int(time.time())),
        )
        self._db.exec("DELETE FROM response_input_items WHERE response_id = ?", (rid,))
        for i, item in enumerate(inputs):
            self._db.exec(
                "INSERT INTO response_input_items (response_id, seq, body) VALUES (?, ?, ?)",
                (rid, i, json.dumps(item)),
            )
        return rid

    async def get_response(self, response_id: str) -> Optional[Dict[str, Any]]:
        rows = self._db.query("SELECT body FROM responses WHERE id = ?", (response_id,))
        return json.loads(rows[0]["body"]) if rows else None

    async def delete_response(self, response_id: str) -> bool:
        self._db.exec("DELETE FROM response_input_items WHERE response_id = ?", (response_id,))
        cur = self._db.exec("DELETE FROM responses WHERE id = ?", (response_id,))
        return cur.rowcount > 0

    async def list_input_items(self, response_id: str) -> List[Dict[str, Any]]:
        rows = self._db.query(
            "SELECT body FROM response_input_items WHERE response_id = ? ORDER BY seq",
            (response_id,),
        )
        return [json.loads(r["body"]) for r in rows]


class SqliteConversationStorage(ConversationStorage):
    def __init__(self, db: _Db):
        self._db = db

    async def create_conversation(self, metadata: Optional[Dict] = None) -> Dict[str, Any]:
        cid = f"conv_{uuid.uuid4().hex}"
        created = int(time.time())
        self._db.exec(
            "INSERT INTO conversations (id, metadata, created_at) VALUES (?, ?, ?)",
            (cid, json.dumps(metadata or {}), created),
        )
        return {"id": cid, "object": "conversation", "created_at": created, "metadata": metadata or {}}

    async def get_conversation(self, conv_id: str) -> Optional[Dict[str, Any]]:
        rows = self._db.query("SELECT * FROM conversations WHERE id = ?", (conv_id,))
        if not rows:
            return None
        r = rows[0]
        return {"id": r["id"], "object": "conversation", "created_at": r["created_at"], "metadata": json.loads(r["metadata"])}

    async def update_conversation(self, conv_id: str, metadata: Dict) -> Optional[Dict[str, Any]]:
        cur = self._db.exec("UPDATE conversations SET metadata = ? WHERE id = ?", (json.dumps(metadata), conv_id))
        if cur.rowcount == 0:
            return None
        return await self.get_conversation(conv_id)

    async def delete_conversation(self, conv_id: str) -> bool:
        self._db.exec("DELETE FROM conversation_items WHERE conversation_id = ?", (conv_id,))
        cur = self._db.exec("DELETE FROM conversations WHERE id = ?", (conv_id,))
        return cur.rowcount > 0

    async def add_items(self, conv_id: str, items: List[Dict]) -> List[Dict[str, Any]]:
        if await self.get_conversation(conv_id) is None:
            raise StorageError(f"conversation {conv_id} not found")
        rows = self._db.query(
            "SELECT COALESCE(MAX(seq), -1) AS m FROM conversation_items WHERE conversation_id = ?",
            (conv_id,),
        )
        seq = rows[0]["m"] + 1
        out = []
        for item in items:
            item = dict(item)
            item.setdefault("id", f"item_{uuid.uuid4().hex}")
            item.setdefault("created_at", int(time.time()))
            self._db.exec(
                "INSERT OR REPLACE INTO conversation_items (conversation_id, item_id, seq, body)"
                " VALUES (?, ?, ?, ?)",
                (conv_id, item["id"], seq, json.dumps(item)),
            )
            seq += 1
            out.append(item)
        return out

    async def list_items(self, conv_id: str, limit: int = 100, after: Optional[str] = None) -> List[Dict]:
        if after:
            rows = self._db.query(
                "SELECT seq FROM conversation_items WHERE conversation_id = ? AND item_id = ?",
                (conv_id, after),
            )
            after_seq = rows[0]["seq"] if rows else -1
        else:
            after_seq = -1
        rows = self._db.query(
            "SELECT body FROM conversation_items WHERE conversation_id = ? AND seq > ?"
            " ORDER BY seq LIMIT ?",
            (conv_id, after_seq, limit),
        )
        return [json.loads(r["body"]) for r in rows]

    async def get_item(self, conv_id: str, item_id: str) -> Optional[Dict]:
        rows = self._db.query(
            "SELECT body FROM conversation_items WHERE conversation_id = ? AND item_id = ?",
            (conv_id, item_id),
        )
        return json.loads(rows[0]["body"]) if rows else None

    async def delete_item(self, conv_id: str, item_id: str) -> bool:
        cur = self._db.exec(
            "DELETE FROM conversation_items WHERE conversation_id = ? AND item_id = ?",
            (conv_id, item_id),
        )
        return cur.rowcount > 0


def open_sqlite(url: str):
    """'sqlite' / 'sqlite://' -> in-memory; 'sqlite:///path' -> file."""
    path = ":memory:"
    if url.startswith("sqlite:///"):
        path = url[len("sqlite:///"):] or ":memory:"
    db = _Db(path)
    return SqliteResponseStorage(db), SqliteConversationStorage(db)
