"""Storage traits + memory/noop backends (reference data_connector/src/lib.rs,
memory.rs, noop.rs)."""
from __future__ import annotations

import time
import uuid
from typing import Any, Dict, List, Optional


class StorageError(RuntimeError):
    pass


class ResponseStorage:
    async def store_response(self, response: Dict[str, Any]) -> str:
        raise NotImplementedError

    async def get_response(self, response_id: str) -> Optional[Dict[str, Any]]:
        raise NotImplementedError

    async def delete_response(self, response_id: str) -> bool:
        raise NotImplementedError

    async def list_input_items(self, response_id: str) -> List[Dict[str, Any]]:
        raise NotImplementedError


class ConversationStorage:
    async def create_conversation(self, metadata: Optional[Dict] = None) -> Dict[str, Any]:
        raise NotImplementedError

    async def get_conversation(self, conv_id: str) -> Optional[Dict[str, Any]]:
        raise NotImplementedError

    async def update_conversation(self, conv_id: str, metadata: Dict) -> Optional[Dict[str, Any]]:
        raise NotImplementedError

    async def delete_conversation(self, conv_id: str) -> bool:
        raise NotImplementedError

    async def add_items(self, conv_id: str, items: List[Dict]) -> List[Dict[str, Any]]:
        raise NotImplementedError

    async def list_items(self, conv_id: str, limit: int = 100, after: Optional[str] = None) -> List[Dict]:
        raise NotImplementedError

    async def get_item(self, conv_id: str, item_id: str) -> Optional[Dict]:
        raise NotImplementedError

    async def delete_item(self, conv_id: str, item_id: str) -> bool:
        raise NotImplementedError


class MemoryResponseStorage(ResponseStorage):
    def __init__(self):
        self._responses: Dict[str, Dict] = {}
        self._inputs: Dict[str, List[Dict]] = {}

    async def store_response(self, response: Dict[str, Any]) -> str:
        rid = response.get("id") or f"resp_{uuid.uuid4().hex}"
        response["id"] = rid
        self._responses[rid] = response
        self._inputs[rid] = response.pop("_input_items", [])
        return rid

    async def get_response(self, response_id: str):
        return self._responses.get(response_id)

    async def delete_response(self, response_id: str) -> bool:
        self._inputs.pop(response_id, None)
        return self._responses.pop(response_id, None) is not None

    async def list_input_items(self, response_id: str):
        return self._inputs.get(response_id, [])


class MemoryConversationStorage(ConversationStorage):
    def __init__(self):
        self._convs: Dict[str, Dict] = {}
        self._items: Dict[str, List[Dict]] = {}

    async def create_conversation(self, metadata=None):
        cid = f"conv_{uuid.uuid4().hex}"
        conv = {"id": cid, "object": "conversation", "created_at": int(time.time()), "metadata": metadata or {}}
        self._convs[cid] = conv
        self._items[cid] = []
        return conv

    async def get_conversation(self, conv_id):
        return self._convs.get(conv_id)

    async def update_conversation(self, conv_id, metadata):
        conv = self._convs.get(conv_id)
        if conv is None:
            return None
        conv["metadata"] = metadata
        return conv

    async def delete_conversation(self, conv_id):
        self._items.pop(conv_id, None)
        return self._convs.pop(conv_id, None) is not None

    async def add_items(self, conv_id, items):
        if conv_id not in self._convs:
            raise StorageError(f"conversation {conv_id} not found")
        out = []
        for item in items:
            item = dict(item)
            item.setdefault("id", f"item_{uuid.uuid4().hex}")
            item.setdefault("created_at", int(time.time()))
            self._items[conv_id].append(item)
            out.append(item)
        return out

    async def list_items(self, conv_id, limit=100, after=None):
        items = self._items.get(conv_id, [])
        if after:
            idx = next((i for i, it in enumerate(items) if it["id"] == after), -1)
            items = items[idx + 1:]
        return items[:limit]

    async def get_item(self, conv_id, item_id):
        for it in self._items.get(conv_id, []):
            if it["id"] == item_id:
                return it
        return None

    async def delete_item(self, conv_id, item_id):
        items = self._items.get(conv_id, [])
        n = len(items)
        self._items[conv_id] = [it for it in items if it["id"] != item_id]
        return len(self._items[conv_id]) < n


class NoopResponseStorage(ResponseStorage):
    async def store_response(self, response):
        return response.get("id") or f"resp_{uuid.uuid4().hex}"

    async def get_response(self, response_id):
        return None

    async def delete_response(self, response_id):
        return False

    async def list_input_items(self, response_id):
        return []


def make_storage(backend: str):
    """-> (ResponseStorage, ConversationStorage)."""
    if backend in ("memory",):
        return MemoryResponseStorage(), MemoryConversationStorage()
    if backend in ("none", "noop"):
        return NoopResponseStorage(), MemoryConversationStorage()
    if backend == "sqlite" or backend.startswith("sqlite:"):
        from .sqlite import open_sqlite

        return open_sqlite(backend)
    if backend == "redis":
        raise StorageError("redis backend requires a redis driver (not in this image)")
    if backend == "postgres":
        raise StorageError("postgres backend requires asyncpg/psycopg (not in this image)")
    if backend == "oracle":
        raise StorageError("oracle backend requires oracledb (not in this image)")
    raise StorageError(f"unknown storage backend {backend!r}")
