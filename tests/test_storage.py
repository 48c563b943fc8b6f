"""Object storage unit tests (reference behavior: storage.py:44-90)."""

import asyncio
import re

import pytest

from code_interpreter_amd.services.storage import Storage


def test_write_read_roundtrip(tmp_path):
    async def run():
        storage = Storage(str(tmp_path / "objs"))
        h = await storage.write(b"hello world")
        assert re.fullmatch(r"[0-9a-f]{64}", h)
        assert await storage.read(h) == b"hello world"
        assert await storage.exists(h)
        assert not await storage.exists("0" * 64)

    asyncio.run(run())


def test_streaming_writer_reader(tmp_path):
    async def run():
        storage = Storage(str(tmp_path / "objs"))
        async with storage.writer() as w:
            await w.write(b"part1-")
            await w.write(b"part2")
            h = w.hash
        async with storage.reader(h) as r:
            assert await r.read() == b"part1-part2"

    asyncio.run(run())


def test_missing_object_raises(tmp_path):
    async def run():
        storage = Storage(str(tmp_path / "objs"))
        with pytest.raises(FileNotFoundError):
            await storage.read("ab" * 32)

    asyncio.run(run())


def test_reader_validates_hash(tmp_path):
    """Path traversal through the hash is rejected by pattern validation."""

    async def run():
        storage = Storage(str(tmp_path / "objs"))
        with pytest.raises(Exception):
            await storage.read("../../etc/passwd")

    asyncio.run(run())
