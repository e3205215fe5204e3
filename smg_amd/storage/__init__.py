"""Chat-history / Responses / Conversations storage (reference:
crates/data_connector — ResponseStorage / ConversationStorage /
ConversationItemStorage traits (lib.rs); backends memory / noop / postgres /
redis / oracle).

This build ships memory, noop and sqlite (SQL persistence, stdlib driver —
`sqlite` in-memory or `sqlite:///path.db` on disk) backends in-tree;
postgres/redis backends instantiate lazily and raise a clear error when
their drivers are absent from the image (no network installs).
"""
from .base import ConversationStorage, ResponseStorage, StorageError, make_storage

__all__ = ["ConversationStorage", "ResponseStorage", "StorageError", "make_storage"]
