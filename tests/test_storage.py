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


def test_storage_gc(tmp_path):
    import os
    import time as time_mod

    from code_interpreter_amd.storage_gc import collect

    async def run():
        storage = Storage(str(tmp_path))
        old = await storage.write(b"old")
        new = await storage.write(b"new")
        return old, new

    old, new = asyncio.run(run())
    past = time_mod.time() - 100_000
    os.utime(tmp_path / old, (past, past))

    stats = collect(str(tmp_path), ttl_hours=1.0, dry_run=True)
    assert stats == {"removed": 1, "kept": 1, "freed_bytes": 3}
    assert (tmp_path / old).exists()

    stats = collect(str(tmp_path), ttl_hours=1.0)
    assert stats["removed"] == 1
    assert not (tmp_path / old).exists()
    assert (tmp_path / new).exists()


def test_storage_gc_ttl(tmp_path):
    import os
    import time

    from code_interpreter_amd.storage_gc import collect

    root = tmp_path / "store"
    root.mkdir()
    old = root / ("a" * 64)
    new = root / ("b" * 64)
    junk = root / "not-an-object.txt"
    for p in (old, new, junk):
        p.write_bytes(b"x" * 100)
    stale = time.time() - 48 * 3600
    os.utime(old, (stale, stale))

    dry = collect(str(root), ttl_hours=24, dry_run=True)
    assert dry["removed"] == 1 and old.exists()

    out = collect(str(root), ttl_hours=24)
    assert out["removed"] == 1 and out["kept"] == 1
    assert not old.exists()
    assert new.exists() and junk.exists()  # non-object names untouched
