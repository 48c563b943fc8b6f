"""Direct unit tests for the small utility modules (they are exercised
indirectly everywhere; these pin their exact contracts)."""

import asyncio

import pytest

from code_interpreter_amd.utils.envsettings import EnvSettings
from code_interpreter_amd.utils.retry import async_retry
from code_interpreter_amd.utils.validation import validate_hash
from code_interpreter_amd.utils import gpus


# -- retry ------------------------------------------------------------------
def test_retry_succeeds_after_failures(monkeypatch):
    sleeps = []

    async def fake_sleep(d):
        sleeps.append(d)

    monkeypatch.setattr(asyncio, "sleep", fake_sleep)
    calls = {"n": 0}

    async def flaky():
        calls["n"] += 1
        if calls["n"] < 3:
            raise RuntimeError("boom")
        return "ok"

    assert asyncio.run(async_retry(flaky, attempts=3)) == "ok"
    assert calls["n"] == 3
    assert len(sleeps) == 2
    # jittered exponential: first in [2,4], second in [4,8]
    assert 2.0 <= sleeps[0] <= 4.0
    assert 4.0 <= sleeps[1] <= 8.0


def test_retry_exhausts_and_reraises(monkeypatch):
    async def fake_sleep(d):
        pass

    monkeypatch.setattr(asyncio, "sleep", fake_sleep)

    async def always_fails():
        raise RuntimeError("nope")

    with pytest.raises(RuntimeError, match="nope"):
        asyncio.run(async_retry(always_fails, attempts=3))


def test_retry_does_not_catch_other_exceptions():
    async def wrong_kind():
        raise ValueError("not retryable")

    with pytest.raises(ValueError):
        asyncio.run(async_retry(wrong_kind, attempts=3, retry_on=(RuntimeError,)))


# -- env settings -----------------------------------------------------------
class _Settings(EnvSettings):
    name: str = "default"
    count: int = 5
    extras: dict = {}


def test_envsettings_reads_prefixed_vars(monkeypatch):
    monkeypatch.setenv("APP_NAME", "fromenv")
    monkeypatch.setenv("APP_COUNT", "42")
    monkeypatch.setenv("APP_EXTRAS", '{"a": 1}')
    s = _Settings()
    assert (s.name, s.count, s.extras) == ("fromenv", 42, {"a": 1})


def test_envsettings_explicit_beats_env(monkeypatch):
    monkeypatch.setenv("APP_COUNT", "42")
    assert _Settings(count=7).count == 7


def test_envsettings_empty_env_ignored(monkeypatch):
    monkeypatch.setenv("APP_NAME", "")
    assert _Settings().name == "default"


def test_envsettings_string_not_json_decoded(monkeypatch):
    # a str-typed field must receive the raw value even if it looks like JSON
    monkeypatch.setenv("APP_NAME", '{"not": "parsed"}')
    assert _Settings().name == '{"not": "parsed"}'


# -- validation -------------------------------------------------------------
def test_validate_hash_accepts_token():
    validate_hash("a" * 64)
    validate_hash("0123abcdEF_-")


@pytest.mark.parametrize("bad", ["", "x" * 256, "../etc/passwd", "a b", "a/b"])
def test_validate_hash_rejects(bad):
    with pytest.raises(ValueError):
        validate_hash(bad)


# -- gpu detection ----------------------------------------------------------
def test_detect_gpu_count_fake_sysfs(tmp_path, monkeypatch):
    gpus.detect_gpu_count.cache_clear()
    kfd = tmp_path / "topology" / "nodes"
    for i, gfx in enumerate(["0", "950", "950"]):  # node 0 = CPU
        node = kfd / str(i)
        node.mkdir(parents=True)
        (node / "properties").write_text(
            f"simd_count {0 if gfx == '0' else 256}\ngfx_target_version {gfx}\n"
        )
    monkeypatch.setattr(gpus, "KFD_NODES", str(kfd))
    monkeypatch.delenv("HIP_VISIBLE_DEVICES", raising=False)
    assert gpus.detect_gpu_count() == 2


def test_detect_gpu_count_respects_visibility(tmp_path, monkeypatch):
    gpus.detect_gpu_count.cache_clear()
    kfd = tmp_path / "topology" / "nodes"
    for i in range(1, 5):
        node = kfd / str(i)
        node.mkdir(parents=True)
        (node / "properties").write_text("simd_count 256\n")
    monkeypatch.setattr(gpus, "KFD_NODES", str(kfd))
    monkeypatch.setenv("HIP_VISIBLE_DEVICES", "0,2")
    assert gpus.detect_gpu_count() == 2
    gpus.detect_gpu_count.cache_clear()
