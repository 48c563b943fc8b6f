"""Flat-file object storage for workspace file round-trips.

Objects are opaque byte blobs named by a random 64-hex token ("hash" in the
wire API; the reference names objects the same way, storage.py:52 — note its
ids are random tokens, not content digests, and ours match that contract so
existing clients' GC/TTL assumptions hold).

API parity: Storage.writer()/reader()/read()/write()/exists()
(reference storage.py:44-90).
"""

import secrets
from contextlib import asynccontextmanager
from typing import AsyncIterator, Protocol

from anyio import Path

from code_interpreter_amd.utils.validation import validate_hash


class ObjectReader(Protocol):
    async def read(self, size: int = -1) -> bytes: ...


class ObjectWriter(Protocol):
    hash: str

    async def write(self, data: bytes) -> None: ...


class Storage:
    def __init__(self, storage_path: str):
        self.storage_path = Path(storage_path)

    @asynccontextmanager
    async def writer(self) -> AsyncIterator[ObjectWriter]:
        """Write a new object; its id is available as ``.hash``."""
        await self.storage_path.mkdir(parents=True, exist_ok=True)
        object_id = secrets.token_hex(32)
        async with await (self.storage_path / object_id).open("wb") as f:
            f.hash = object_id  # type: ignore[attr-defined]
            yield f

    @asynccontextmanager
    async def reader(self, object_hash: str) -> AsyncIterator[ObjectReader]:
        validate_hash(object_hash)
        target = self.storage_path / object_hash
        if not object_hash or not await target.exists():
            raise FileNotFoundError(f"File not found: {object_hash}")
        async with await target.open("rb") as f:
            yield f

    async def write(self, data: bytes) -> str:
        async with self.writer() as f:
            await f.write(data)
            return f.hash

    async def read(self, object_hash: str) -> bytes:
        async with self.reader(object_hash) as f:
            return await f.read()

    async def exists(self, object_hash: str) -> bool:
        validate_hash(object_hash)
        return await (self.storage_path / object_hash).exists()
